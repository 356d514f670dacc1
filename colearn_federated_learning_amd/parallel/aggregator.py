"""Round aggregation: weighted all-reduce of model deltas over RCCL/xGMI.

SURVEY.md §2b N2/N3.  The reference aggregates by shipping every client's
full state dict to the coordinator over websockets and averaging in Python
 — O(clients x model) bytes through one host.  Here
each rank pre-scales its clients' summed delta by their aggregate sample
weight, one all-reduce(SUM) combines both the deltas and the scalar weights,
and every rank divides — mathematically identical weighted FedAvg, but
bandwidth-optimal on the 8-GPU xGMI mesh.

xGMI-aware chunking: each MI355X GPU has 7 point-to-point links (~153 GB/s
each); a single monolithic ring all-reduce is bound by ONE link.  The flat
delta is therefore all-reduced as a stream of bucket_bytes chunks on
round-robin RCCL streams (separate process-group communicators), which lets
RCCL schedule multiple rings over distinct links.  Bucket size is a config
knob to be swept on hardware (2-16 MB).

The gloo backend (CPU plumbing config) reuses the same code path minus
streams.  Single-process runs skip communication entirely.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from ..config import AggregationConfig
from ..ops.topk import topk_decompress


class Aggregator:
    def __init__(
        self,
        cfg: AggregationConfig,
        device: torch.device,
    ) -> None:
        self.cfg = cfg
        self.device = device
        self.world = dist.get_world_size() if dist.is_initialized() else 1
        self._groups: List[object] = []
        self._streams: List[torch.cuda.Stream] = []
        if self.world > 1 and device.type == "cuda" and cfg.comm_streams > 0:
            # Optional: several communicators let bucket all-reduces overlap
            # so RCCL can drive distinct xGMI links concurrently.  Off by
            # default — a single large RCCL all-reduce already uses multiple
            # channels, and this path can't be validated on 1-GPU boxes.
            for _ in range(cfg.comm_streams):
                self._groups.append(dist.new_group(backend=dist.get_backend()))
                self._streams.append(torch.cuda.Stream(device=device))

    # ------------------------------------------------------------------
    def allreduce_weighted(
        self, flat_delta: torch.Tensor, weight: float
    ) -> Tuple[torch.Tensor, float]:
        """In-place: flat_delta <- sum_ranks(flat_delta), returns total weight.

        Caller pre-scales its local contribution by ``weight`` before the
        call (weighted mean = scaled SUM / total weight — SURVEY.md §5c).
        """
        if self.world == 1:
            return flat_delta, weight
        wt = torch.tensor([weight], dtype=torch.float64)
        if dist.get_backend() != "gloo":
            wt = wt.to(self.device)
        dist.all_reduce(wt, op=dist.ReduceOp.SUM)
        total_weight = float(wt.item())

        if self.cfg.topk_ratio > 0 and self.cfg.transport == "sparse":
            self._sparse_allreduce(flat_delta)
        else:
            self._dense_allreduce(flat_delta)
        return flat_delta, total_weight

    # ------------------------------------------------------------------
    @staticmethod
    def _staged(t: torch.Tensor) -> bool:
        """gloo can't reduce CUDA tensors: stage through host.  This lets
        the whole multi-process GPU engine run over gloo transport — the
        1-GPU-box validation path for multi-rank training (RCCL refuses
        two ranks on one device: 'Duplicate GPU detected')."""
        return t.is_cuda and dist.get_backend() == "gloo"

    def _allreduce_inplace(self, t: torch.Tensor) -> None:
        if self._staged(t):
            c = t.cpu()
            dist.all_reduce(c, op=dist.ReduceOp.SUM)
            t.copy_(c)
        else:
            dist.all_reduce(t, op=dist.ReduceOp.SUM)

    def _dense_allreduce(self, flat: torch.Tensor) -> None:
        bucket_elems = max(self.cfg.bucket_bytes // flat.element_size(), 1)
        if not self._streams or flat.numel() <= bucket_elems:
            self._allreduce_inplace(flat)
            return
        works = []
        n = flat.numel()
        current = torch.cuda.current_stream(self.device)
        for i, start in enumerate(range(0, n, bucket_elems)):
            chunk = flat[start : start + bucket_elems]
            stream = self._streams[i % len(self._streams)]
            group = self._groups[i % len(self._groups)]
            stream.wait_stream(current)
            with torch.cuda.stream(stream):
                works.append(
                    dist.all_reduce(chunk, op=dist.ReduceOp.SUM, group=group,
                                    async_op=True)
                )
        for w in works:
            w.wait()
        for s in self._streams:
            current.wait_stream(s)

    # ------------------------------------------------------------------
    def _sparse_allreduce(self, flat: torch.Tensor) -> None:
        """Sparse transport: all-gather (idx, val) nonzeros + scatter-add.

        Sparse deltas don't all-reduce natively.  After
        per-client top-k compression the rank accumulator holds the union
        of its clients' kept entries; this path ships exactly those
        nonzeros — LOSSLESS, so it produces the same global model as the
        dense all-reduce (tests/test_equivalence.py) while moving
        world x nnz x 8 B instead of numel x 4 B.  Counts are equalized to
        the max across ranks with (idx=0, val=0) padding so the collective
        stays fixed-size.
        """
        nz = flat.nonzero(as_tuple=False).reshape(-1)
        k = torch.tensor([nz.numel()], dtype=torch.int64)
        if dist.get_backend() != "gloo":
            k = k.to(self.device)
        dist.all_reduce(k, op=dist.ReduceOp.MAX)
        kmax = max(int(k.item()), 1)
        idx = torch.zeros(kmax, dtype=torch.int32, device=flat.device)
        val = torch.zeros(kmax, dtype=torch.float32, device=flat.device)
        idx[: nz.numel()] = nz.to(torch.int32)
        val[: nz.numel()] = flat[nz].float()
        if dist.get_backend() == "gloo":
            if idx.is_cuda:  # host-staged (see _staged)
                idx, val = idx.cpu(), val.cpu()
            idx_list = [torch.empty_like(idx) for _ in range(self.world)]
            val_list = [torch.empty_like(val) for _ in range(self.world)]
            dist.all_gather(idx_list, idx)
            dist.all_gather(val_list, val)
            if flat.is_cuda:
                idx_list = [i.to(flat.device) for i in idx_list]
                val_list = [v.to(flat.device) for v in val_list]
        else:
            idx_all = torch.empty(self.world * kmax, dtype=idx.dtype,
                                  device=idx.device)
            val_all = torch.empty(self.world * kmax, dtype=val.dtype,
                                  device=val.device)
            dist.all_gather_into_tensor(idx_all, idx)
            dist.all_gather_into_tensor(val_all, val)
            idx_list = list(idx_all.chunk(self.world))
            val_list = list(val_all.chunk(self.world))
        flat.zero_()
        for i, v in zip(idx_list, val_list):
            topk_decompress(i, v, flat.numel(), out=flat)

    # ------------------------------------------------------------------
    def broadcast_flat(self, flat: torch.Tensor, src: int = 0) -> None:
        if self.world > 1:
            if self._staged(flat):
                c = flat.cpu()
                dist.broadcast(c, src=src)
                flat.copy_(c)
            else:
                dist.broadcast(flat, src=src)

    def allreduce_tensor(self, t: torch.Tensor) -> None:
        if self.world > 1 and t.numel():
            self._allreduce_inplace(t)

    def allreduce_scalar(self, value: float, op=None) -> float:
        if self.world == 1:
            return value
        t = torch.tensor([value], dtype=torch.float64)
        if dist.get_backend() != "gloo":
            t = t.to(self.device)
        dist.all_reduce(t, op=op or dist.ReduceOp.SUM)
        return float(t.item())
