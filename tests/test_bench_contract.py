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
