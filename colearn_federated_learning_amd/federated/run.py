"""Top-level entry: ``colearn.run(config)``.

Works single-process (1 device, all virtual clients) or under torchrun
(one rank per GPU over RCCL).  This is the new build's equivalent of the
reference's coordinator script main().
"""

from __future__ import annotations

from typing import Dict, List

import torch.distributed as dist

from ..config import FederatedConfig
from ..utils.dist import init_distributed
from .coordinator import FederatedServer


def run(cfg: FederatedConfig) -> List[Dict[str, float]]:
    rank, world, device = init_distributed(cfg.agg.backend)
    server = FederatedServer(cfg, rank=rank, world=world, device=device)
    history = server.train()
    if dist.is_initialized():
        dist.barrier()
    return history


def main() -> None:
    import argparse

    from ..config import driver_config, load_yaml, parse_overrides

    ap = argparse.ArgumentParser(description="CoLearn-AMD federated run")
    ap.add_argument("--config", type=str, default=None,
                    help="YAML config path")
    ap.add_argument("--driver-config", type=int, default=None,
                    help="BASELINE.json config index 1..5")
    ap.add_argument("overrides", nargs="*",
                    help="dotted key=value overrides, e.g. train.lr=0.1")
    args = ap.parse_args()
    if args.driver_config is not None:
        cfg = driver_config(args.driver_config)
    elif args.config:
        cfg = load_yaml(args.config)
    else:
        cfg = FederatedConfig()
    if args.overrides:
        cfg = cfg.override(**parse_overrides(args.overrides))
    history = run(cfg)
    if history:
        last = history[-1]
        print(f"final round {last['round']}: "
              f"loss={last['mean_client_loss']:.4f} "
              f"samples/s={last['samples_per_sec']:.1f}")


if __name__ == "__main__":
    main()
