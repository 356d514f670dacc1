"""xGMI bucket-size x comm-streams sweep for the weighted delta all-reduce
(SURVEY.md §5 comm backend, §7 Phase 3).

Each MI355X GPU has 7 point-to-point xGMI links (~153 GB/s each); a single
monolithic ring all-reduce is bound by ONE link, so the aggregator can chunk
the flat delta across several communicators/streams.  This harness measures
exactly that trade: algorithmic bus bandwidth of the aggregator's dense
all-reduce over a grid of (bucket_bytes, comm_streams) at the launched world
size, for ResNet-18/50-sized deltas.

Launch (the driver's 8-GPU box, or any N):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 tools/comm_sweep.py \
        [--sizes 11.7e6,25.6e6] [--out gpurun_out/comm_sweep.json]

Also runs on CPU/gloo (plumbing check; bandwidth numbers meaningless).
Rank 0 writes one JSON with GB/s per grid point and the argmax.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist

# repo root on sys.path when launched by file path (torchrun)
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def bench_point(flat: torch.Tensor, bucket_bytes: int, streams: int,
                device: torch.device, iters: int = 10) -> float:
    """Returns seconds per all-reduce (MAX over ranks)."""
    from colearn_federated_learning_amd.config import AggregationConfig
    from colearn_federated_learning_amd.parallel.aggregator import Aggregator

    cfg = AggregationConfig(bucket_bytes=bucket_bytes, comm_streams=streams)
    agg = Aggregator(cfg, device)
    for _ in range(3):
        agg._dense_allreduce(flat)
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        agg._dense_allreduce(flat)
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    elapsed = (time.perf_counter() - t0) / iters
    t = torch.tensor([elapsed], dtype=torch.float64)
    if dist.get_backend() != "gloo":
        t = t.to(device)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--sizes", type=str, default="11.7e6,25.6e6",
                    help="flat delta sizes in elements (fp32); defaults = "
                         "ResNet-18 and ResNet-50 param counts")
    ap.add_argument("--buckets", type=str,
                    default="1,2,4,8,16,32,64",
                    help="bucket sizes in MiB (0 = monolithic)")
    ap.add_argument("--streams", type=str, default="0,2,4,7",
                    help="comm_streams values (0 = single communicator)")
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--out", type=str, default="gpurun_out/comm_sweep.json")
    args = ap.parse_args()

    from colearn_federated_learning_amd.utils.dist import init_distributed

    rank, world, device = init_distributed("auto")
    if world == 1:
        print("world=1: nothing to sweep (launch under torchrun)")
        return
    results = []
    for size in [int(float(s)) for s in args.sizes.split(",")]:
        flat = torch.randn(size, dtype=torch.float32, device=device)
        nbytes = size * 4
        for streams in [int(s) for s in args.streams.split(",")]:
            buckets = ([0] if streams == 0
                       else [int(b) for b in args.buckets.split(",")])
            for mib in buckets:
                bucket_bytes = (1 << 62) if mib == 0 else mib * (1 << 20)
                sec = bench_point(flat, bucket_bytes, streams, device,
                                  args.iters)
                # ring all-reduce algo bandwidth: 2(N-1)/N * bytes / time
                algbw = nbytes / sec / 1e9
                busbw = algbw * 2 * (world - 1) / world
                rec = {"elements": size, "bucket_mib": mib,
                       "comm_streams": streams, "sec": sec,
                       "algbw_GBs": round(algbw, 2),
                       "busbw_GBs": round(busbw, 2)}
                results.append(rec)
                if rank == 0:
                    print(json.dumps(rec), flush=True)
    if rank == 0:
        best = max(results, key=lambda r: r["busbw_GBs"])
        out = {"world": world, "device": str(device),
               "results": results, "best": best}
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            json.dump(out, f, indent=1)
        print("BEST", json.dumps(best))
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
