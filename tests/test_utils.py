"""Utility-layer tests (SURVEY.md layer 9)."""

import json

import numpy as np
import torch

from gossipy_amd.utils import (
    StringEncoder,
    choice_not_n,
    plot_evaluation,
    torch_models_eq,
)


def test_choice_not_n_never_returns_excluded():
    for _ in range(200):
        v = choice_not_n(0, 10, 5)
        assert 0 <= v <= 10 and v != 5


def test_torch_models_eq():
    a = torch.nn.Linear(4, 2)
    b = torch.nn.Linear(4, 2)
    assert not torch_models_eq(a, b)
    b.load_state_dict(a.state_dict())
    assert torch_models_eq(a, b)


def test_string_encoder_falls_back_to_str():
    class Odd:
        def __str__(self):
            return "odd!"

    out = json.dumps({"x": Odd()}, cls=StringEncoder)
    assert "odd!" in out


def test_plot_evaluation_headless(tmp_path, monkeypatch):
    import matplotlib

    matplotlib.use("Agg")
    # both accepted shapes: plain dicts and the report's (round, dict)
    plot_evaluation(
        [[{"accuracy": 0.5}, {"accuracy": 0.7}]], title="dicts"
    )
    plot_evaluation(
        [[(0, {"accuracy": 0.5}), (1, {"accuracy": 0.7})]], title="tuples"
    )
