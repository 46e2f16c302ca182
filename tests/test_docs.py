"""The generated API reference must stay generatable: every documented
module imports and renders (guards against docstring/introspection rot)."""

import subprocess
import sys

import pytest


@pytest.mark.timeout(300)
def test_api_docs_generate(tmp_path):
    r = subprocess.run(
        [sys.executable, "docs/gen_api.py"],
        capture_output=True,
        text=True,
        timeout=280,
    )
    assert r.returncode == 0, r.stderr
    assert "gossipy_amd_engine_runner.md" in r.stdout
