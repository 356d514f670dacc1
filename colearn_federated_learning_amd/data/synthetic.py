"""Synthetic datasets with the shapes of the driver configs.

The reference partitions real torchvision datasets across clients
(SURVEY.md §2a row 8); this environment has no network, so every run uses
synthetic data of the named shape with deterministic per-class structure:
inputs for class ``c`` are drawn from N(mu_c, I) with class-dependent mean
direction, which makes the learning problem real (loss decreases, accuracy
rises above chance) while remaining fully reproducible from the seed.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Iterator, Tuple

import torch

from ..utils.seed import stable_hash

from ..config import DataConfig, ModelConfig


@dataclass
class ClientShard:
    """One client's local dataset, fully materialized (small shards)."""

    x: torch.Tensor  # images: [N,C,H,W] fp32 | text: token ids [N,S] int64
    y: torch.Tensor  # labels [N] int64  | MLM: labels [N,S] (-100 = unmasked)
    client_id: int

    def __len__(self) -> int:
        return self.x.shape[0]

    def batches(
        self, batch_size: int, *, generator: torch.Generator = None,
        drop_last: bool = False,
    ) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        n = len(self)
        perm = (
            torch.randperm(n, generator=generator)
            if generator is not None else torch.arange(n)
        )
        end = n - n % batch_size if drop_last else n
        for i in range(0, end, batch_size):
            idx = perm[i : i + batch_size]
            yield self.x[idx], self.y[idx]


def _class_means(num_classes: int, dim: int, gen: torch.Generator) -> torch.Tensor:
    means = torch.randn(num_classes, dim, generator=gen)
    return 1.5 * means / means.norm(dim=1, keepdim=True)


def make_client_dataset(
    data_cfg: DataConfig, model_cfg: ModelConfig, client_id: int,
    labels: torch.Tensor,
) -> ClientShard:
    """Materialize one client's shard for the given label sequence.

    ``labels`` comes from the sharder (IID or Dirichlet) so that the label
    distribution — not the conditional input distribution — is what varies
    across clients, exactly the non-IID axis the reference's partitioner
    controls.
    """
    if data_cfg.dataset == "file":
        from .files import load_file_dataset

        x, y = load_file_dataset(data_cfg.data_path)
        idx = labels  # the sharder returned sample indices for "file"
        return ClientShard(x[idx].clone(), y[idx].clone(), client_id)
    n = labels.shape[0]
    gen = torch.Generator().manual_seed(
        stable_hash(data_cfg.seed, "inputs", client_id)
    )
    if data_cfg.dataset in ("mnist_synth", "cifar10_synth"):
        c = model_cfg.in_channels
        h = w = model_cfg.image_size
        dim = c * h * w
        mean_gen = torch.Generator().manual_seed(data_cfg.seed)
        means = _class_means(data_cfg.num_classes, dim, mean_gen)
        x = torch.randn(n, dim, generator=gen) * 0.5 + means[labels]
        return ClientShard(x.view(n, c, h, w).contiguous(), labels, client_id)
    if data_cfg.dataset == "text_synth":
        s = data_cfg.seq_len
        v = data_cfg.vocab_size
        if v < 104:
            raise ValueError(
                f"text_synth needs vocab_size >= 104 (the [MASK]-style "
                f"token id is 103), got {v}")
        lo = min(1000, v // 2)  # reserve low ids for special tokens
        ids = torch.randint(lo, v, (n, s), generator=gen)
        # MLM masking: select ~mlm_probability positions; label = original id
        mask = torch.rand(n, s, generator=gen) < data_cfg.mlm_probability
        mask[:, 0] = False  # keep a CLS-like position intact
        mlm_labels = torch.full((n, s), -100, dtype=torch.long)
        mlm_labels[mask] = ids[mask]
        ids = ids.clone()
        ids[mask] = 103  # [MASK]-style token id
        return ClientShard(ids, mlm_labels, client_id)
    raise ValueError(f"unknown dataset {data_cfg.dataset!r}")
