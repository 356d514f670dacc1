"""The examples/ scripts must keep running (user-facing API surface)."""

import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.parametrize("script", [
    "run_federated.py", "dp_federated.py", "bert_mlm.py",
    "file_dataset.py",
])
def test_example_runs(script):
    out = subprocess.run(
        [sys.executable, os.path.join("examples", script)],
        cwd=ROOT, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "round" in out.stdout
