"""Shared pytest configuration.

Registers the ``gpu`` marker: tests that need a real MI355X are marked
``@pytest.mark.gpu`` and are skipped on CPU-only machines; everything else
must pass without a GPU.
"""

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a GPU (run on the MI355X box)"
    )


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU on this machine")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(autouse=True)
def _clear_global_cache():
    """Each test starts with an empty model cache."""
    from gossipy_amd import CACHE

    CACHE.clear()
    yield
    CACHE.clear()
