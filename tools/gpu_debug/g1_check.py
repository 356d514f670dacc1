"""G=1 stacked-engine GPU check.

The 8-GPU driver layout puts ONE client per rank (configs 2-4 have 8
clients), so every stacked kernel runs with G=1 there — a path never
exercised by the 1-GPU benches (G=8).  A [1,H] grouped-LN weight grad
lost its slot dim on this path (fixed in ops/norms.py); this probe runs
one short round of each stacked family at num_clients=1 on the GPU so
any remaining G=1 shape/kernel bug surfaces before the driver's scaling
tier.  Run: python tools/gpu_debug/g1_check.py
"""

import sys
import os

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch  # noqa: E402

from colearn_federated_learning_amd import driver_config, run  # noqa: E402


def main() -> None:
    assert torch.cuda.is_available()
    # argv = stack sizes to exercise (default 1).  The driver's scaling
    # tiers at N=8/4/2 give G=1/2/4 per rank for the 8-client configs.
    gs = [int(a) for a in sys.argv[1:]] or [1]
    for g in gs:
        for idx in (2, 4, 5):  # resnet18, bert_base, DP-resnet18
            cfg = driver_config(idx).override(
                rounds=1, num_clients=g, num_evaluators=0,
                data__samples_per_client=256,
                train__max_local_steps=2,
                checkpoint_every=0, checkpoint_dir=f"/tmp/g{g}_{idx}",
            )
            hist = run(cfg)
            loss = hist[-1]["mean_client_loss"]
            assert loss == loss, f"config {idx}: NaN loss at G={g}"
            print(f"config {idx} G={g} OK loss={loss:.4f}", flush=True)
    print("G_CHECK_PASS")


if __name__ == "__main__":
    main()
