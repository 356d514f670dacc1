"""Benchmark harness (driver contract).

Flagship: the BASELINE.json headline config — ResNet-18 CIFAR-10-shape
FedAvg, 8 IID clients, bf16, synthetic data, random-init weights.  One
bench "step" is one federated ROUND: every one of the 8 clients runs
local SGD over its shard (local_epochs=1) on its owning GPU, then the
round aggregates via the weighted RCCL all-reduce over xGMI.  The metric
is whole-node samples/sec: (samples trained per round x rounds) / elapsed.

Scaling is STRONG: the 8 clients are fixed (that is the named config) and
are multiplexed over however many GPUs the driver gives us — 8 clients on
1 GPU at N=1 down to 1 client per GPU at N=8.

Launch:  python bench.py --gpus N --steps K --warmup W
(for N>1 the driver launches via torch.distributed.run, one rank per GPU.)
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--config", type=int, default=2,
                    help="BASELINE.json driver config index (default: "
                         "flagship ResNet-18 FedAvg)")
    ap.add_argument("--samples-per-client", type=int, default=None,
                    help="override shard size (default: the driver "
                         "config's own value)")
    ap.add_argument("--set", action="append", default=[],
                    help="config overrides, dotted key=value")
    ap.add_argument("--target-accuracy", type=float, default=None,
                    help="also run a rounds-to-target measurement after "
                         "the throughput steps (eval per round, early stop)")
    args = ap.parse_args()

    from colearn_federated_learning_amd.config import driver_config
    from colearn_federated_learning_amd.federated.coordinator import (
        FederatedServer,
    )
    from colearn_federated_learning_amd.utils.dist import init_distributed

    from colearn_federated_learning_amd.config import parse_overrides

    cfg = driver_config(args.config)
    on_gpu = torch.cuda.is_available()
    cfg = cfg.override(
        rounds=args.warmup + args.steps,
        train__dtype="bfloat16" if on_gpu else "float32",
        num_evaluators=0,
        checkpoint_dir=None,
        metrics_path=None,
    )
    if args.samples_per_client is not None:
        cfg = cfg.override(
            data__samples_per_client=args.samples_per_client)
    if args.set:
        cfg = cfg.override(**parse_overrides(args.set))
    rank, world, device = init_distributed(cfg.agg.backend)
    server = FederatedServer(cfg, rank=rank, world=world, device=device)

    def barrier_sync():
        if dist.is_initialized():
            dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    for r in range(args.warmup):
        server.run_round(r)
    barrier_sync()
    t0 = time.perf_counter()
    total_samples = 0.0
    for r in range(args.warmup, args.warmup + args.steps):
        rec = server.run_round(r)
        total_samples += rec["n_samples"]
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # optional companion metric from BASELINE.json: rounds-to-target.
    # Runs the engine's eval + early-stop loop on a fresh server (the
    # throughput run above has no evaluators in its timed rounds).
    rounds_to_target = None
    if args.target_accuracy is not None:
        cfg_t = cfg.override(rounds=200, num_evaluators=1,
                             target_accuracy=args.target_accuracy)
        server_t = FederatedServer(cfg_t, rank=rank, world=world,
                                   device=device)
        hist = server_t.train()
        if hist and hist[-1].get("eval_accuracy", 0) >= args.target_accuracy:
            rounds_to_target = len(hist)

    # MAX elapsed over ranks (the job is as slow as its slowest rank)
    if dist.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64)
        if dist.get_backend() != "gloo":
            t = t.to(device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    metric_names = {
        1: "samples/sec (whole node), logreg FedAvg 2 clients",
        2: "samples/sec (whole node), ResNet-18 FedAvg 8 clients",
        3: "samples/sec (whole node), ResNet-50 FedProx 8 clients",
        4: "samples/sec (whole node), BERT-base MLM FedAvg 8 clients",
        5: "samples/sec (whole node), DP-FedAvg+top-k ResNet-18 64 clients",
    }
    if rank == 0:
        value = total_samples / elapsed
        result = {
            "metric": metric_names[args.config],
            "value": value,
            "unit": "samples/sec",
            "n_gpus": args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            # the driver configs fix TOTAL work (num_clients and shard
            # sizes don't grow with N): adding GPUs splits the same
            # clients across ranks => strong scaling
            "scaling": "strong",
            "vs_baseline": None,
            "rounds_to_target": rounds_to_target,
            "target_accuracy": args.target_accuracy,
            "dtype": "bf16" if on_gpu else "float32",
            "data": "synthetic",
            "config": {
                "model": cfg.model.name,
                "driver_config": args.config,
                "global_batch": cfg.train.batch_size * cfg.num_clients,
                "local_batch": cfg.train.batch_size,
                "num_clients": cfg.num_clients,
                "samples_per_client": cfg.data.samples_per_client,
                "image": f"{cfg.model.in_channels}x{cfg.model.image_size}"
                         f"x{cfg.model.image_size}",
                "parallelism": f"federated dp{args.gpus}",
                "local_epochs": cfg.train.local_epochs,
            },
        }
        print(json.dumps(result))

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
