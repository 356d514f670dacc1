"""bench.py driver-contract guard: one tiny CPU run must emit exactly one
JSON line with the schema the benchmark driver parses."""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "1",
         "--warmup", "0", "--samples-per-client", "64",
         "--set", "num_clients=2", "--set", "train__batch_size=16"],
        cwd=ROOT, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    required = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"}
    assert required <= set(d), sorted(required - set(d))
    assert d["n_gpus"] == 1 and d["steps"] == 1 and d["warmup"] == 0
    assert d["higher_is_better"] is True
    assert d["scaling"] in ("weak", "strong")
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["data"] == "synthetic"
    assert "ResNet-18" in d["metric"]
    assert d["config"]["model"] == "resnet18"


def test_bench_world2_torchrun():
    """The driver's exact N>1 launch shape (torch.distributed.run with 2
    ranks over gloo on CPU): rank 0 must emit one JSON line with
    n_gpus=2 and the whole-job aggregate value."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), "bench.py", "--gpus", "2",
         "--steps", "1", "--warmup", "0", "--samples-per-client", "32",
         "--set", "num_clients=2", "--set", "train__batch_size=16"],
        cwd=ROOT, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.strip().splitlines()
             if ln.startswith("{")]
    assert len(lines) == 1, out.stdout[-2000:]  # ONE line, rank 0 only
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2 and d["value"] > 0
    assert "dp2" in d["config"]["parallelism"]
