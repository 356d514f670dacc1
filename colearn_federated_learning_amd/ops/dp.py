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
    pre_scale: float = 1.0, groups: int = 1,
) -> List[torch.Tensor]:
    """Clip each sample's total grad to clip_norm; return per-param means.

    ``pre_scale``: the per-sample grads are treated as pre_scale * g (the
    hook engine passes unscaled grads of a MEAN loss with pre_scale = B)
    — the factor folds into the clip scale so no extra sweep is needed.

    ``groups`` > 1 (stacked DP): the B rows are ``groups`` clients'
    samples client-major; returns per-param per-CLIENT means
    [groups, *shape] — the whole multi-client DP step reduces in one
    launch per parameter.
    """
    b = per_sample[0].shape[0]
    norms = per_sample_norms(per_sample) * pre_scale
    scale = ((clip_norm / norms.clamp_min(1e-12)).clamp_max(1.0)
             * pre_scale)  # [B]
    out = []
    if per_sample[0].is_cuda and native_enabled():
        m = require_ext()
        for g in per_sample:
            shape = ((groups,) + tuple(g.shape[1:]) if groups > 1
                     else g.shape[1:])
            o = torch.empty(shape, device=g.device, dtype=torch.float32)
            m.scale_rows_mean(g.reshape(b, -1), scale, o.reshape(-1),
                              groups)
            out.append(o)
        return out
    for g in per_sample:
        flat = (g.reshape(b, -1).float() * scale[:, None])
        if groups > 1:
            o = flat.view(groups, b // groups, -1).mean(dim=1)
            o = o.reshape((groups,) + tuple(g.shape[1:]))
        else:
            o = flat.mean(dim=0).reshape(g.shape[1:])
        out.append(o)
    return out


def add_dp_noise(
    flat_delta: torch.Tensor, clip_norm: float, noise_multiplier: float,
    generator: torch.Generator = None, key: int = None,
) -> None:
    """In-place Gaussian mechanism on the aggregated (summed) delta.

    Keyed ONLY on ``key`` (derived from (seed, round)) so the noise vector
    is identical on every rank layout (1 GPU vs 8 GPUs) — part of the
    simulation-equivalence contract.  On gfx950 the noise is generated
    device-side by a philox4x32-10 + Box-Muller HIP kernel (deterministic
    per element index, no host randn + H2D round trip); the CPU path draws
    from a torch CPU generator.  The two paths produce different (equally
    valid) Gaussian samples.
    """
    if noise_multiplier <= 0:
        return
    std = noise_multiplier * clip_norm
    if flat_delta.is_cuda and native_enabled():
        assert key is not None or generator is not None
        if key is None:
            key = int(generator.initial_seed())
        require_ext().philox_gaussian_add(flat_delta, key, std)
        return
    if generator is None and key is not None:
        generator = torch.Generator().manual_seed(key)
    noise = torch.randn(flat_delta.shape, generator=generator,
                        dtype=torch.float32)
    flat_delta.add_(noise.to(flat_delta.device), alpha=std)


class PrivacyAccountant:
    """RDP accountant for the subsampled Gaussian mechanism.

    Per-step Renyi DP at integer order alpha (Mironov, Talwar, Zhang 2019,
    "Renyi Differential Privacy of the Sampled Gaussian Mechanism", Thm 5 /
    eq. (3) binomial expansion — valid upper bound for integer alpha >= 2):

        RDP(alpha) = 1/(alpha-1) * log( sum_{k=0..alpha} C(alpha,k)
                       (1-q)^(alpha-k) q^k exp((k^2-k)/(2 sigma^2)) )

    with the q=1 case reducing to the plain Gaussian alpha/(2 sigma^2).
    Composition is additive over steps; conversion to (eps, delta) uses the
    improved bound (Balle et al. 2020, as in Opacus):

        eps = rdp - (log(delta) + log(alpha))/(alpha-1) + log((alpha-1)/alpha)

    minimized over a grid of integer orders.  Reporting-grade.
    """

    ORDERS = list(range(2, 65)) + [80, 96, 128, 192, 256, 512]

    def __init__(self, noise_multiplier: float, sample_rate: float,
                 delta: float = 1e-5) -> None:
        self.sigma = noise_multiplier
        self.q = sample_rate
        self.delta = delta
        self.steps = 0
        self._rdp_step = None  # per-order cache (sigma/q fixed per run)

    def step(self, n: int = 1) -> None:
        self.steps += n

    # -- per-step RDP at integer order alpha ---------------------------
    def _rdp_one(self, alpha: int) -> float:
        import math

        if self.q >= 1.0:
            return alpha / (2 * self.sigma ** 2)
        if self.q <= 0.0:
            return 0.0
        log_q = math.log(self.q)
        log_1q = math.log1p(-self.q)
        # log-sum-exp over k of:
        #   log C(alpha,k) + k log q + (alpha-k) log(1-q) + (k^2-k)/(2 s^2)
        terms = []
        for k in range(alpha + 1):
            log_binom = (math.lgamma(alpha + 1) - math.lgamma(k + 1)
                         - math.lgamma(alpha - k + 1))
            terms.append(log_binom + k * log_q + (alpha - k) * log_1q
                         + (k * k - k) / (2 * self.sigma ** 2))
        mx = max(terms)
        lse = mx + math.log(sum(math.exp(t - mx) for t in terms))
        return max(lse / (alpha - 1), 0.0)

    def epsilon(self) -> float:
        if self.sigma <= 0 or self.steps == 0:
            return float("inf")
        import math

        if self._rdp_step is None:
            self._rdp_step = [self._rdp_one(a) for a in self.ORDERS]
        best = float("inf")
        for alpha, r in zip(self.ORDERS, self._rdp_step):
            rdp = self.steps * r
            eps = (rdp - (math.log(self.delta) + math.log(alpha))
                   / (alpha - 1) + math.log((alpha - 1) / alpha))
            best = min(best, eps)
        return best
