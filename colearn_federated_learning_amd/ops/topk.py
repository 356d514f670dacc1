"""Top-k gradient sparsification (index+value packing).

SURVEY.md §2b N10: keep the k largest-|value| entries of the flat client
delta; the dense all-reduce is replaced by an all-gather of (idx, val)
pairs + local scatter-add (xGMI analysis in SURVEY.md §5: sparse deltas
don't all-reduce naturally).  The gfx950 kernel (csrc/dp_topk.hip) does a
threshold-estimation pass (histogram over |v| exponents) + compaction;
CPU path uses torch.topk (oracle).
"""

from __future__ import annotations

from typing import Tuple

import torch

from . import native_enabled, require_ext


def topk_compress(
    flat: torch.Tensor, k: int
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Return (indices int32 [k], values fp32 [k]) of the k largest |entries|."""
    k = min(k, flat.numel())
    if flat.is_cuda and native_enabled():
        return require_ext().topk_compress(flat.reshape(-1), k)
    _, idx = torch.topk(flat.abs().reshape(-1), k, sorted=False)
    idx, _ = idx.sort()
    return idx.to(torch.int32), flat.reshape(-1)[idx].float()


def topk_decompress(
    idx: torch.Tensor, val: torch.Tensor, numel: int,
    out: torch.Tensor = None,
) -> torch.Tensor:
    """Scatter-add (idx, val) into a dense flat tensor."""
    if out is None:
        out = torch.zeros(numel, dtype=torch.float32, device=val.device)
    if out.is_cuda and native_enabled():
        require_ext().scatter_add_pairs(out, idx, val)
        return out
    out.index_add_(0, idx.long(), val.to(out.dtype))
    return out
