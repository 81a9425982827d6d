"""The examples/ scripts stay runnable (docs-rot guard)."""

import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.parametrize("script", [
    "quickstart.py", "vector_search.py", "flight_client.py",
    "cdc_ingest.py",
])
def test_example_runs(script):
    if script == "flight_client.py":
        pytest.importorskip("pyarrow.flight")
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "examples", script)],
        capture_output=True, text=True, timeout=600,
        env=dict(os.environ, LAKESOUL_TIMING="0"),
    )
    assert r.returncode == 0, r.stderr[-1500:]
    assert "OK" in r.stdout
