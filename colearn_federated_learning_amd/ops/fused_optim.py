"""Fused SGD / Adam optimizers (multi-tensor-apply style, one launch).

SURVEY.md §2b N8: on gfx950 the whole optimizer step for all parameters runs
as a single HIP kernel over a flat view of the parameter/grad/momentum
buffers — pure HBM bandwidth, fused with the FedProx proximal term
(N14: g += mu * (w - w_global)) so the proximal pass costs no extra memory
sweep.  CPU tensors use an equivalent torch._foreach_* implementation which
doubles as the numerics oracle for tests/test_kernels_gpu.py.

Flat-buffer layout: parameters are trained out of ONE contiguous fp32 buffer
(see parallel.flat.FlatParams), so the fused kernels take single tensors,
not tensor lists — simpler, and the same flat buffer feeds the bucketed
RCCL all-reduce aggregator with zero repacking.
"""

from __future__ import annotations

from typing import Optional

import torch

from . import native_enabled, require_ext


class FlatSGD:
    """SGD(momentum, weight-decay, FedProx mu) over one flat parameter buffer.

    update: g = grad + wd * w + mu * (w - w_ref)
            m = momentum * m + g
            w = w - lr * m
    """

    def __init__(
        self,
        flat_params: torch.Tensor,
        flat_grads: torch.Tensor,
        lr: float,
        momentum: float = 0.0,
        weight_decay: float = 0.0,
        fedprox_mu: float = 0.0,
        flat_ref: Optional[torch.Tensor] = None,
    ) -> None:
        assert flat_params.shape == flat_grads.shape
        self.w = flat_params
        self.g = flat_grads
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.fedprox_mu = fedprox_mu
        self.ref = flat_ref
        self.m = (
            torch.zeros_like(flat_params) if momentum != 0.0 else None
        )

    def reset(self) -> None:
        if self.m is not None:
            self.m.zero_()

    def host_prepare(self) -> None:
        pass  # SGD's update is stateless on the host: graph-safe as-is

    @torch.no_grad()
    def device_step(self) -> None:
        mu = self.fedprox_mu
        require_ext().fused_sgd(
            self.w, self.g,
            self.m if self.m is not None
            else torch.empty(0, device=self.w.device),
            self.ref if (mu != 0.0 and self.ref is not None)
            else torch.empty(0, device=self.w.device),
            self.lr, self.momentum, self.weight_decay, mu,
        )

    @torch.no_grad()
    def step(self) -> None:
        mu = self.fedprox_mu
        if self.w.is_cuda and native_enabled():
            self.device_step()
            return
        g = self.g
        if self.weight_decay != 0.0:
            g = g.add(self.w, alpha=self.weight_decay)
        if mu != 0.0 and self.ref is not None:
            g = g.add(self.w - self.ref, alpha=mu)
        if self.m is not None:
            self.m.mul_(self.momentum).add_(g)
            g = self.m
        self.w.add_(g, alpha=-self.lr)

    def zero_grad(self) -> None:
        self.g.zero_()


class FlatAdam:
    """Adam(W) over one flat parameter buffer, with optional FedProx term."""

    def __init__(
        self,
        flat_params: torch.Tensor,
        flat_grads: torch.Tensor,
        lr: float,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
        fedprox_mu: float = 0.0,
        flat_ref: Optional[torch.Tensor] = None,
    ) -> None:
        self.w = flat_params
        self.g = flat_grads
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.fedprox_mu = fedprox_mu
        self.ref = flat_ref
        self.m = torch.zeros_like(flat_params)
        self.v = torch.zeros_like(flat_params)
        self.t = 0
        # live bias-correction buffer: under hipGraph replay the kernel's
        # scalar args are frozen, so [bc1, bc2] travels through device
        # memory, refreshed by host_prepare() before each step/replay.
        self._bc = (
            torch.ones(2, device=flat_params.device)
            if flat_params.is_cuda else None
        )

    def reset(self) -> None:
        self.m.zero_()
        self.v.zero_()
        self.t = 0

    def host_prepare(self) -> None:
        """Host-side half of step(): advance t, refresh the device bc
        buffer.  Must run OUTSIDE any graph capture/replay."""
        self.t += 1
        if self._bc is not None:
            bc = torch.tensor(
                [1.0 - self.beta1 ** self.t, 1.0 - self.beta2 ** self.t],
                dtype=torch.float32,
            )
            self._bc.copy_(bc, non_blocking=True)

    @torch.no_grad()
    def device_step(self) -> None:
        """Device-side half (graph-capturable)."""
        mu = self.fedprox_mu
        require_ext().fused_adam(
            self.w, self.g, self.m, self.v,
            self.ref if (mu != 0.0 and self.ref is not None)
            else torch.empty(0, device=self.w.device),
            self._bc,
            self.lr, self.beta1, self.beta2, self.eps,
            self.weight_decay, mu, 1.0, 1.0,
        )

    @torch.no_grad()
    def step(self) -> None:
        if self.w.is_cuda and native_enabled():
            self.host_prepare()
            self.device_step()
            return
        self.t += 1
        bc1 = 1.0 - self.beta1 ** self.t
        bc2 = 1.0 - self.beta2 ** self.t
        mu = self.fedprox_mu
        g = self.g
        if self.weight_decay != 0.0:
            g = g.add(self.w, alpha=self.weight_decay)
        if mu != 0.0 and self.ref is not None:
            g = g.add(self.w - self.ref, alpha=mu)
        self.m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
        self.v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
        denom = (self.v / bc2).sqrt_().add_(self.eps)
        self.w.addcdiv_(self.m, denom, value=-self.lr / bc1)

    def zero_grad(self) -> None:
        self.g.zero_()


def make_flat_optimizer(kind: str, flat_params, flat_grads, train_cfg,
                        flat_ref=None):
    if kind == "sgd":
        return FlatSGD(
            flat_params, flat_grads, lr=train_cfg.lr,
            momentum=train_cfg.momentum,
            weight_decay=train_cfg.weight_decay,
            fedprox_mu=train_cfg.fedprox_mu, flat_ref=flat_ref,
        )
    if kind == "adam":
        return FlatAdam(
            flat_params, flat_grads, lr=train_cfg.lr,
            weight_decay=train_cfg.weight_decay,
            fedprox_mu=train_cfg.fedprox_mu, flat_ref=flat_ref,
        )
    raise ValueError(f"unknown optimizer {kind!r}")
