"""Client / server entry points (reference parity: SURVEY.md §2a rows 1-2).

The reference deploys FL as separate processes: each device runs a client
worker script (PySyft websocket server + MQTT availability announcement),
and a coordinator script drives sessions against them.  This engine is
SPMD — every process runs the same round loop and aggregation is a
collective — so the deployment mapping is:

  reference                      | this engine
  -------------------------------+------------------------------------
  MQTT broker (discovery)        | TCP rendezvous (MASTER_ADDR/PORT)
  client worker process          | ``colearn client --rank K``
  coordinator process            | ``colearn server`` (rank 0)
  websocket model up/download    | weighted all-reduce (RCCL/gloo)
  availability/role/session msgs | MessageBus topics (per process)

``server`` with ``--world 1`` (the default) runs the whole federation in
one process — the single-node simulation mode.

Usage:
  python -m colearn_federated_learning_amd.cli server --driver-config 2
  python -m colearn_federated_learning_amd.cli client --rank 1 --world 4 \\
      --master-addr 10.0.0.1 --driver-config 2
"""

from __future__ import annotations

import argparse
import os
import sys
from typing import List, Optional


def _add_common(ap: argparse.ArgumentParser) -> None:
    ap.add_argument("--config", type=str, default=None, help="YAML config")
    ap.add_argument("--driver-config", type=int, default=None,
                    help="BASELINE.json config index 1..5")
    ap.add_argument("--master-addr", type=str, default="127.0.0.1")
    ap.add_argument("--master-port", type=int, default=29711)
    ap.add_argument("--world", type=int, default=1,
                    help="total participant processes")
    ap.add_argument("overrides", nargs="*",
                    help="dotted key=value overrides, e.g. train.lr=0.1")


def _build_cfg(args):
    from .config import FederatedConfig, driver_config, load_yaml, \
        parse_overrides

    if args.driver_config is not None:
        cfg = driver_config(args.driver_config)
    elif args.config:
        cfg = load_yaml(args.config)
    else:
        cfg = FederatedConfig()
    if args.overrides:
        cfg = cfg.override(**parse_overrides(args.overrides))
    return cfg


def _launch(rank: int, args) -> None:
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(args.world)
    os.environ.setdefault("LOCAL_RANK", str(rank))
    os.environ["MASTER_ADDR"] = args.master_addr
    os.environ["MASTER_PORT"] = str(args.master_port)
    from .federated.run import run

    cfg = _build_cfg(args)
    history = run(cfg)
    import torch.distributed as dist

    if dist.is_initialized():
        # exit with the group destroyed: a live gloo group at interpreter
        # shutdown can SIGABRT in its background thread
        dist.destroy_process_group()
    if rank == 0 and history:
        last = history[-1]
        print(f"session done: round {last['round']} "
              f"loss={last['mean_client_loss']:.4f} "
              f"samples/s={last['samples_per_sec']:.1f}")


def main(argv: Optional[List[str]] = None) -> None:
    ap = argparse.ArgumentParser(prog="colearn")
    sub = ap.add_subparsers(dest="cmd", required=True)

    aps = sub.add_parser("server", help="coordinator (rank 0)")
    _add_common(aps)

    apc = sub.add_parser("client", help="client worker (rank >= 1)")
    apc.add_argument("--rank", type=int, required=True,
                     help="this client's rank (1..world-1)")
    _add_common(apc)

    args = ap.parse_args(argv)
    if args.cmd == "server":
        _launch(0, args)
    else:
        if not (1 <= args.rank < args.world):
            sys.exit("client --rank must be in [1, world)")
        _launch(args.rank, args)


if __name__ == "__main__":
    main()
