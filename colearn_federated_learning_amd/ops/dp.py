"""DP-FedAvg primitives: per-sample grad clipping + Gaussian noise.

SURVEY.md §2b N9.  Per-sample gradients (shape [B, *param]) are clipped to
total L2 norm <= C per sample, then averaged over the batch; Gaussian noise
sigma*C is added once per round to the *aggregate* client delta (DP-FedAvg,
McMahan et al. 2018-style).  On gfx950 the norm reduction + clip-scale +
batch mean run as fused HIP kernels over a flat [B, D] view
(csrc/dp_topk.hip); CPU path is the torch oracle.
"""

from __future__ import annotations

from typing import List, Sequence

import torch

from . import native_enabled, require_ext


def per_sample_norms(per_sample: Sequence[torch.Tensor]) -> torch.Tensor:
    """Total L2 norm per sample across all parameter grads. [B]"""
    b = per_sample[0].shape[0]
    if per_sample[0].is_cuda and native_enabled():
        m = require_ext()
        sq = torch.zeros(b, device=per_sample[0].device, dtype=torch.float32)
        for g in per_sample:
            m.sq_norm_per_row(g.reshape(b, -1), sq)  # sq += rowwise |g|^2
        return sq.sqrt()
    sq = torch.zeros(b, dtype=torch.float32)
    for g in per_sample:
        sq += g.reshape(b, -1).float().pow(2).sum(dim=1)
    return sq.sqrt()


def clip_and_reduce(
    per_sample: Sequence[torch.Tensor], clip_norm: float,
    pre_scale: float = 1.0,
) -> List[torch.Tensor]:
    """Clip each sample's total grad to clip_norm; return per-param means.

    ``pre_scale``: the per-sample grads are treated as pre_scale * g (the
    hook engine passes unscaled grads of a MEAN loss with pre_scale = B)
    — the factor folds into the clip scale so no extra sweep is needed.
    """
    b = per_sample[0].shape[0]
    norms = per_sample_norms(per_sample) * pre_scale
    scale = ((clip_norm / norms.clamp_min(1e-12)).clamp_max(1.0)
             * pre_scale)  # [B]
    out = []
    if per_sample[0].is_cuda and native_enabled():
        m = require_ext()
        for g in per_sample:
            o = torch.empty(g.shape[1:], device=g.device, dtype=torch.float32)
            m.scale_rows_mean(g.reshape(b, -1), scale, o.reshape(-1))
            out.append(o)
        return out
    for g in per_sample:
        flat = g.reshape(b, -1).float()
        o = (flat * scale[:, None]).mean(dim=0).reshape(g.shape[1:])
        out.append(o)
    return out


def add_dp_noise(
    flat_delta: torch.Tensor, clip_norm: float, noise_multiplier: float,
    generator: torch.Generator = None,
) -> None:
    """In-place Gaussian mechanism on the aggregated (summed) delta."""
    if noise_multiplier <= 0:
        return
    std = noise_multiplier * clip_norm
    # noise is drawn on the CPU generator so it is identical for every rank
    # layout (1 GPU vs 8 GPUs) — part of the simulation-equivalence contract.
    noise = torch.randn(flat_delta.shape, generator=generator,
                        dtype=torch.float32)
    flat_delta.add_(noise.to(flat_delta.device), alpha=std)


class PrivacyAccountant:
    """Moments-accountant-lite: tracks (epochs of) noisy rounds and reports
    an RDP-based (epsilon, delta) estimate for the Gaussian mechanism.

    For sampled Gaussian with sampling rate q and noise sigma over T rounds,
    uses the standard RDP of the Gaussian mechanism composed T times,
    converted at the given delta.  Suitable for reporting, not certification.
    """

    def __init__(self, noise_multiplier: float, sample_rate: float,
                 delta: float = 1e-5) -> None:
        self.sigma = noise_multiplier
        self.q = sample_rate
        self.delta = delta
        self.steps = 0

    def step(self, n: int = 1) -> None:
        self.steps += n

    def epsilon(self) -> float:
        if self.sigma <= 0 or self.steps == 0:
            return float("inf")
        import math

        best = float("inf")
        for alpha in [1 + x / 10.0 for x in range(1, 1000)]:
            # RDP of subsampled Gaussian (upper bound via q^2 alpha / sigma^2
            # for small q — standard approximation) composed over steps.
            rdp = self.steps * (self.q ** 2) * alpha / (2 * self.sigma ** 2)
            eps = rdp + math.log(1 / self.delta) / (alpha - 1)
            best = min(best, eps)
        return best
