"""Every example script must run end-to-end at a small scale (the
reference's CI-equivalent: the entry scripts are the integration tests,
SURVEY.md §4)."""

import subprocess
import sys

import pytest

ARGS = {
    "examples/main_ormandi_2013.py": ["--nodes", "20", "--rounds", "3"],
    "examples/main_hegedus_2021.py": ["--nodes", "30", "--rounds", "3"],
    "examples/main_giaretta_2019.py": ["--nodes", "30", "--rounds", "3"],
    "examples/main_hegedus_2020.py": [
        "--users", "25", "--items", "50", "--rounds", "3"
    ],
    "examples/main_berta_2014.py": ["--nodes", "10", "--rounds", "3"],
    "examples/main_all2all.py": ["--nodes", "10", "--rounds", "3"],
    "examples/main_danner_2023.py": ["--nodes", "10", "--rounds", "2"],
    "examples/main_onoszko_2021.py": ["--nodes", "8", "--rounds", "3"],
    "examples/baseline.py": ["--samples", "400", "--epochs", "3"],
}


@pytest.mark.parametrize("script", sorted(ARGS))
@pytest.mark.timeout(300)
def test_example_runs(script):
    r = subprocess.run(
        [sys.executable, script, *ARGS[script]],
        capture_output=True,
        text=True,
        timeout=280,
    )
    assert r.returncode == 0, f"{script} failed:\n{r.stdout}\n{r.stderr}"


@pytest.mark.parametrize(
    "extra",
    [
        ["--engine"],
        ["--engine-cnn"],
        ["--engine-cnn", "--pens"],
    ],
    ids=["engine", "engine-cnn", "engine-cnn-pens"],
)
@pytest.mark.timeout(600)
def test_onoszko_engine_variants(extra):
    """The Onoszko example's engine paths (incl. the paper's PENS+CNN
    pairing) run end-to-end at a small scale."""
    r = subprocess.run(
        [sys.executable, "examples/main_onoszko_2021.py",
         "--nodes", "8", "--rounds", "3", *extra],
        capture_output=True,
        text=True,
        timeout=560,
    )
    assert r.returncode == 0, f"{extra} failed:\n{r.stdout}\n{r.stderr}"
