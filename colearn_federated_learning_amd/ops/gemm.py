"""Linear / GEMM with hand-written gfx950 MFMA kernels on the GPU path.

SURVEY.md §2b N5.  Forward, grad-input and grad-weight are three GEMM
flavors (NT, NN, TN) over bf16 operands with fp32 accumulation, implemented
in csrc/gemm_bf16.hip as LDS-staged MFMA kernels (``global_load_lds`` direct
staging, ``v_mfma_f32_16x16x32_bf16`` tiles, XCD-aware block swizzle).
CPU tensors run the stock PyTorch implementation, which is also the test
oracle.  Functions use the setup_context style so torch.func transforms
(vmap/grad — the DP per-sample path) compose with them.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from . import is_vmapped, native_enabled, require_ext


def _use_native(*ts: torch.Tensor) -> bool:
    return (ts[0].is_cuda and native_enabled() and not is_vmapped(*ts)
            and all(t.dtype == torch.bfloat16 for t in ts))


class _LinearFn(torch.autograd.Function):
    generate_vmap_rule = True

    @staticmethod
    def forward(x: torch.Tensor, w: torch.Tensor,
                bias: Optional[torch.Tensor]):
        if _use_native(x, w):
            return require_ext().linear_nt(
                x.contiguous(), w.contiguous(),
                bias if bias is not None
                else torch.empty(0, device=x.device),
            )
        return F.linear(x, w, bias)

    @staticmethod
    def setup_context(ctx, inputs, output):
        x, w, bias = inputs
        ctx.save_for_backward(x, w)
        ctx.has_bias = bias is not None

    @staticmethod
    def backward(ctx, gy: torch.Tensor):
        x, w = ctx.saved_tensors
        gy = gy.contiguous()
        if _use_native(gy, w, x):
            m = require_ext()
            gx = m.gemm_nn(gy, w.contiguous())          # [M,N]@[N,K] -> [M,K]
            gw = m.gemm_tn(gy, x.contiguous())          # [M,N]^T@[M,K]->[N,K]
        else:
            gx = gy @ w
            gw = gy.transpose(-1, -2) @ x
        gb = gy.sum(dim=0) if ctx.has_bias else None
        return gx, gw, gb


def colearn_linear(
    x: torch.Tensor, w: torch.Tensor, bias: Optional[torch.Tensor] = None
) -> torch.Tensor:
    """``x @ w^T + bias``; >2-D inputs are flattened over leading dims.

    Honors autocast: under a bf16 autocast region the operands are cast to
    bf16 and the native MFMA kernel runs.
    """
    if torch.is_autocast_enabled("cuda") and x.is_cuda:
        dt = torch.get_autocast_dtype("cuda")
        x, w = x.to(dt), w.to(dt)
        bias = bias.to(dt) if bias is not None else None
    lead = x.shape[:-1]
    x2 = x.reshape(-1, x.shape[-1])
    out = _LinearFn.apply(x2, w, bias)
    return out.reshape(*lead, w.shape[0])


class _MatmulFn(torch.autograd.Function):
    generate_vmap_rule = True

    @staticmethod
    def forward(a: torch.Tensor, b: torch.Tensor):
        if _use_native(a, b):
            return require_ext().bmm_nn(a.contiguous(), b.contiguous())
        return a @ b

    @staticmethod
    def setup_context(ctx, inputs, output):
        ctx.save_for_backward(*inputs)

    @staticmethod
    def backward(ctx, gy: torch.Tensor):
        a, b = ctx.saved_tensors
        gy = gy.contiguous()
        if _use_native(gy, a, b):
            m = require_ext()
            ga = m.bmm_nt(gy, b.contiguous())           # gy @ b^T
            gb = m.bmm_tn(a.contiguous(), gy)           # a^T @ gy
        else:
            ga = gy @ b.transpose(-1, -2)
            gb = a.transpose(-1, -2) @ gy
        return ga, gb


def colearn_matmul(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Batched ``a @ b`` (attention scores/context); native bf16 batched
    MFMA GEMM on the GPU, torch.matmul elsewhere (oracle)."""
    if torch.is_autocast_enabled("cuda") and a.is_cuda:
        dt = torch.get_autocast_dtype("cuda")
        a, b = a.to(dt), b.to(dt)
    return _MatmulFn.apply(a, b)


class _MatmulNTFn(torch.autograd.Function):
    """a @ b^T for 3-D operands with both K-major (attention scores:
    q [B,S,dh] x k [B,S,dh] -> [B,S,S]) — no transpose copies anywhere."""

    generate_vmap_rule = True

    @staticmethod
    def forward(a: torch.Tensor, b: torch.Tensor):
        if _use_native(a, b):
            return require_ext().bmm_nt(a.contiguous(), b.contiguous())
        return a @ b.transpose(-1, -2)

    @staticmethod
    def setup_context(ctx, inputs, output):
        ctx.save_for_backward(*inputs)

    @staticmethod
    def backward(ctx, gy: torch.Tensor):
        a, b = ctx.saved_tensors
        gy = gy.contiguous()
        if _use_native(gy, a, b):
            m = require_ext()
            ga = m.bmm_nn(gy, b.contiguous())   # gy @ b
            gb = m.bmm_tn(gy, a.contiguous())   # gy^T @ a
        else:
            ga = gy @ b
            gb = gy.transpose(-1, -2) @ a
        return ga, gb


def colearn_matmul_nt(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Batched ``a @ b^T`` (both operands contraction-major)."""
    if torch.is_autocast_enabled("cuda") and a.is_cuda:
        dt = torch.get_autocast_dtype("cuda")
        a, b = a.to(dt), b.to(dt)
    return _MatmulNTFn.apply(a, b)
