"""Real-dataset loading + federated partitioning from a local file.

The reference downloads MNIST via torchvision and partitions it across
clients.  This environment has no network, so the
equivalent capability here is ``dataset: "file"``: a ``torch.save``d dict
``{"x": float tensor [N,C,H,W], "y": int64 [N]}`` (or an (x, y) tuple) is
partitioned across clients with the same IID / Dirichlet(alpha) label-skew
axes as the synthetic sharder — on REAL samples, by index assignment.

Determinism contract matches data/sharder.py: the partition depends only
on (seed, num_clients, client_id, file labels), never on rank layout.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch

from ..config import DataConfig
from ..utils.seed import stable_hash

_CACHE: Dict[str, Tuple[torch.Tensor, torch.Tensor]] = {}


def load_file_dataset(path: str) -> Tuple[torch.Tensor, torch.Tensor]:
    """Load (x, y) from a torch.save'd dict {"x":..., "y":...} or tuple."""
    if path not in _CACHE:
        obj = torch.load(path, map_location="cpu", weights_only=True)
        if isinstance(obj, dict):
            x, y = obj["x"], obj["y"]
        else:
            x, y = obj
        _CACHE[path] = (x.float(), y.long())
    return _CACHE[path]


def partition_file_indices(
    cfg: DataConfig, num_clients: int
) -> List[torch.Tensor]:
    """Per-client sample indices into the file dataset.

    IID: one global seeded shuffle, equal contiguous splits (equal shard
    sizes keep the stacked engine applicable).  Dirichlet: per-client
    class proportions ~ Dir(alpha); samples are drawn class-by-class
    without replacement from seeded per-class pools, so every sample is
    used at most once.
    """
    _, y = load_file_dataset(cfg.data_path)
    n = y.shape[0]
    per = (cfg.samples_per_client if cfg.samples_per_client > 0
           else n // num_clients)
    if per * num_clients > n:
        raise ValueError(
            f"file dataset has {n} samples < "
            f"{num_clients} clients x {per} samples_per_client")
    if cfg.partition == "iid":
        g = torch.Generator().manual_seed(stable_hash(cfg.seed, "file_iid"))
        perm = torch.randperm(n, generator=g)
        return [perm[i * per : (i + 1) * per] for i in range(num_clients)]
    if cfg.partition != "dirichlet":
        raise ValueError(f"unknown partition {cfg.partition!r}")
    # per-class index pools, shuffled once with the global seed
    g = torch.Generator().manual_seed(stable_hash(cfg.seed, "file_dir"))
    classes = sorted(int(c) for c in torch.unique(y))
    pools = {}
    for c in classes:
        idx = (y == c).nonzero(as_tuple=False).reshape(-1)
        pools[c] = idx[torch.randperm(idx.numel(), generator=g)].tolist()
    out = []
    for cid in range(num_clients):
        gc = torch.Generator().manual_seed(
            stable_hash(cfg.seed, "file_dirichlet", cid))
        conc = torch.full((len(classes),), float(cfg.dirichlet_alpha))
        gamma = torch._standard_gamma(conc, generator=gc)
        probs = gamma / gamma.sum().clamp_min(1e-12)
        want = torch.multinomial(probs, per, replacement=True,
                                 generator=gc)
        counts = torch.bincount(want, minlength=len(classes))
        picked: List[int] = []
        for ci, c in enumerate(classes):
            take = min(int(counts[ci]), len(pools[c]))
            picked.extend(pools[c][:take])
            del pools[c][:take]
        # pool exhaustion (heavy skew late clients): fill from any class
        short = per - len(picked)
        if short > 0:
            for c in classes:
                take = min(short, len(pools[c]))
                picked.extend(pools[c][:take])
                del pools[c][:take]
                short -= take
                if short == 0:
                    break
        out.append(torch.tensor(picked[:per], dtype=torch.long))
    return out
