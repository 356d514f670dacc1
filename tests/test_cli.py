"""CLI entry points: client/server processes over localhost rendezvous."""

import json
import os
import subprocess
import sys

import torch


def _run_proc(cmd, env_extra=None):
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env.pop("LOCAL_RANK", None)
    if env_extra:
        env.update(env_extra)
    return subprocess.Popen(
        cmd, env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True,
    )


def test_server_standalone(tmp_path):
    ck = tmp_path / "ck"
    p = _run_proc([
        sys.executable, "-m", "colearn_federated_learning_amd.cli",
        "server", "--driver-config", "1",
        "rounds=2", "data.samples_per_client=64",
        f"checkpoint_dir={ck}", "checkpoint_every=1",
    ])
    out, _ = p.communicate(timeout=300)
    assert p.returncode == 0, out
    assert "session done" in out
    assert (ck / "colearn_latest.pt").exists()


def test_server_plus_client_processes(tmp_path):
    """Two-process federation: coordinator + one client worker, localhost
    rendezvous (the reference's deployment shape)."""
    common = [
        "--driver-config", "1", "--world", "2",
        "--master-port", "29751",
        "rounds=2", "num_clients=2", "data.samples_per_client=64",
        f"checkpoint_dir={tmp_path / 'ck'}",
    ]
    srv = _run_proc([
        sys.executable, "-m", "colearn_federated_learning_amd.cli",
        "server", *common,
    ])
    cli = _run_proc([
        sys.executable, "-m", "colearn_federated_learning_amd.cli",
        "client", "--rank", "1", *common,
    ])
    out_s, _ = srv.communicate(timeout=300)
    out_c, _ = cli.communicate(timeout=300)
    assert srv.returncode == 0, out_s
    assert cli.returncode == 0, out_c
    assert "session done" in out_s
    # and the 2-process session equals the standalone one
    ck = torch.load(tmp_path / "ck" / "colearn_latest.pt",
                    map_location="cpu", weights_only=False)
    assert ck["round"] == 1


def test_module_entry_point(tmp_path):
    """python -m colearn_federated_learning_amd (no .cli suffix) works."""
    p = _run_proc([
        sys.executable, "-m", "colearn_federated_learning_amd",
        "server", "--driver-config", "1",
        "rounds=1", "data.samples_per_client=32",
        f"checkpoint_dir={tmp_path / 'ck'}",
    ])
    out, _ = p.communicate(timeout=300)
    assert p.returncode == 0, out
    assert "session done" in out


def test_gpu_tools_importable():
    """The multi-GPU tools must at least parse/import on CPU so they
    don't rot before the day an 8-GPU node runs them."""
    import importlib.util as ilu

    for rel in ("tools/comm_sweep.py", "tools/rccl_validate.py",
                "tools/gpu_debug/g1_check.py"):
        path = os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), rel)
        spec = ilu.spec_from_file_location("_tool_mod", path)
        mod = ilu.module_from_spec(spec)
        # import-time side effects only; main() guarded by __main__
        spec.loader.exec_module(mod)
        assert hasattr(mod, "main")
