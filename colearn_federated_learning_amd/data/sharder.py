"""Federated data sharding: IID and non-IID Dirichlet(alpha) label skew.

The reference partitions a dataset across clients before training
(SURVEY.md §2a row 8).  Here sharding operates on *label sequences*: each
client receives a deterministic sequence of labels, and the synthetic data
generator materializes inputs conditioned on those labels.  The Dirichlet
scheme is the standard non-IID benchmark partition (per-client class
proportions p_c ~ Dir(alpha)); small alpha => heavily skewed shards.

Determinism contract: shards depend only on (seed, num_clients, client_id),
never on the rank layout, so an N-client run produces identical shards on
1 GPU and on 8 GPUs (test: tests/test_sharder.py).
"""

from __future__ import annotations

from typing import List

import torch

from ..utils.seed import stable_hash

from ..config import DataConfig


def iid_shards(cfg: DataConfig, num_clients: int) -> List[torch.Tensor]:
    """Uniform label draws per client."""
    out = []
    for cid in range(num_clients):
        gen = torch.Generator().manual_seed(
            stable_hash(cfg.seed, "iid", cid)
        )
        out.append(
            torch.randint(0, cfg.num_classes, (cfg.samples_per_client,),
                          generator=gen)
        )
    return out


def dirichlet_shards(cfg: DataConfig, num_clients: int) -> List[torch.Tensor]:
    """Per-client class proportions drawn from Dirichlet(alpha)."""
    out = []
    for cid in range(num_clients):
        gen = torch.Generator().manual_seed(
            stable_hash(cfg.seed, "dirichlet", cid)
        )
        conc = torch.full((cfg.num_classes,), float(cfg.dirichlet_alpha))
        # Gamma draws normalized => Dirichlet sample
        gamma = torch._standard_gamma(conc, generator=gen)
        probs = gamma / gamma.sum().clamp_min(1e-12)
        labels = torch.multinomial(
            probs, cfg.samples_per_client, replacement=True, generator=gen
        )
        out.append(labels)
    return out


def make_shards(cfg: DataConfig, num_clients: int) -> List[torch.Tensor]:
    if cfg.dataset == "file":
        # real-dataset path: "shards" are per-client sample INDICES into
        # the file (data/files.py); make_client_dataset slices by them
        from .files import partition_file_indices

        return partition_file_indices(cfg, num_clients)
    if cfg.dataset == "text_synth":
        # text shards have no label skew axis; each client gets a seeded id
        # sequence of the right length (labels unused, generator keys on cid)
        return [torch.zeros(cfg.samples_per_client, dtype=torch.long)
                for _ in range(num_clients)]
    if cfg.partition == "iid":
        return iid_shards(cfg, num_clients)
    if cfg.partition == "dirichlet":
        return dirichlet_shards(cfg, num_clients)
    raise ValueError(f"unknown partition {cfg.partition!r}")
