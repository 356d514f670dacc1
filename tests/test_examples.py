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


def test_client_server_shell_example():
    """The client/server deployment example (reference shape) runs."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    env = dict(os.environ, PORT=str(port))
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK"):
        env.pop(k, None)
    out = subprocess.run(
        ["bash", os.path.join("examples", "client_server.sh")],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "session done" in out.stdout
