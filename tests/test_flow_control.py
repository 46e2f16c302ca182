"""Unit tests for the token-account flow-control strategies."""

import numpy as np
import pytest

from gossipy_amd import set_seed
from gossipy_amd.flow_control import (
    GeneralizedTokenAccount,
    PurelyProactiveTokenAccount,
    PurelyReactiveTokenAccount,
    RandomizedTokenAccount,
    SimpleTokenAccount,
)


def test_add_sub_floor():
    a = SimpleTokenAccount(C=2)
    a.add(3)
    assert a.n_tokens == 3
    a.sub(5)
    assert a.n_tokens == 0, "balance floors at zero"


def test_purely_proactive():
    a = PurelyProactiveTokenAccount()
    assert a.proactive() == 1
    assert a.reactive(10) == 0


def test_purely_reactive():
    a = PurelyReactiveTokenAccount(k=3)
    assert a.proactive() == 0
    assert a.reactive(2) == 6


def test_simple_account_thresholds():
    a = SimpleTokenAccount(C=2)
    assert a.proactive() == 0 and a.reactive(1) == 0
    a.add(1)
    assert a.proactive() == 0 and a.reactive(1) == 1
    a.add(1)
    assert a.proactive() == 1


def test_generalized_reactive_formula():
    a = GeneralizedTokenAccount(C=10, A=4)
    a.add(9)
    # (A - 1 + tokens) / A = (4-1+9)/4 = 3
    assert a.reactive(1) == 3
    # halved without utility
    assert a.reactive(0) == 1


def test_generalized_validates_args():
    with pytest.raises(AssertionError):
        GeneralizedTokenAccount(C=2, A=5)


def test_randomized_proactive_ramp():
    a = RandomizedTokenAccount(C=20, A=10)
    assert a.proactive() == 0
    a.n_tokens = 9
    assert a.proactive() == 0.0
    a.n_tokens = 20
    assert a.proactive() == 1
    a.n_tokens = 15
    assert 0 < a.proactive() < 1
    a.n_tokens = 25
    assert a.proactive() == 1


def test_randomized_reactive_rounding():
    set_seed(0)
    a = RandomizedTokenAccount(C=20, A=10)
    a.n_tokens = 25  # r = 2.5
    draws = [a.reactive(1) for _ in range(300)]
    assert set(draws) <= {2, 3}
    assert abs(np.mean(draws) - 2.5) < 0.15
    assert a.reactive(0) == 0
