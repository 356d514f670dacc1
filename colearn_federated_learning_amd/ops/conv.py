"""2-D convolution with implicit-GEMM HIP kernels on the GPU path.

SURVEY.md §2b N4: ResNet-18/50's 3x3 and 1x1 convs in bf16 as
implicit-GEMM MFMA kernels (fwd, dgrad, wgrad) in csrc/conv_bf16.hip.
CPU path = F.conv2d (oracle).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from . import is_vmapped, native_enabled, require_ext


def _use_native(*ts: torch.Tensor) -> bool:
    return (ts[0].is_cuda and native_enabled() and not is_vmapped(*ts)
            and all(t.dtype == torch.bfloat16 for t in ts))


class _Conv2dFn(torch.autograd.Function):
    """Weights are K-MAJOR: [CO, KH, KW, CI].  This is the conv kernels'
    native operand layout, so the model path runs with ZERO per-call weight
    permute passes (fwd: none; dgrad: one for the transposed view; wgrad:
    none — the fp32 slab output IS [CO, R, S, CI])."""

    generate_vmap_rule = True

    @staticmethod
    def forward(x, w, stride: int, padding: int):
        if _use_native(x, w):
            # layout normalization (channels-last) happens in the binding
            return require_ext().conv2d_fwd_km(x, w, stride, padding)
        return F.conv2d(x, w.permute(0, 3, 1, 2), None, stride, padding)

    @staticmethod
    def setup_context(ctx, inputs, output):
        x, w, stride, padding = inputs
        ctx.save_for_backward(x, w)
        ctx.stride, ctx.padding = stride, padding

    @staticmethod
    def backward(ctx, gy):
        x, w = ctx.saved_tensors
        s, p = ctx.stride, ctx.padding
        need_gx = ctx.needs_input_grad[0]  # stem: input needs no grad
        if _use_native(gy, w, x):
            m = require_ext()
            gx = (m.conv2d_dgrad_km(gy, w, x.shape[2], x.shape[3], s, p)
                  if need_gx else None)
            gw = m.conv2d_wgrad_km(gy, x, w.shape[1], w.shape[2], s, p, 1)
        else:
            ws = w.permute(0, 3, 1, 2)
            gx = (torch.nn.grad.conv2d_input(x.shape, ws, gy, s, p)
                  if need_gx else None)
            gw = torch.nn.grad.conv2d_weight(x, ws.shape, gy, s, p)
            gw = gw.permute(0, 2, 3, 1)
        return gx, gw, None, None


def colearn_conv2d(
    x: torch.Tensor, w: torch.Tensor, stride: int = 1, padding: int = 0
) -> torch.Tensor:
    """w is K-major [CO, KH, KW, CI] (see _Conv2dFn)."""
    if torch.is_autocast_enabled("cuda") and x.is_cuda:
        dt = torch.get_autocast_dtype("cuda")
        x, w = x.to(dt), w.to(dt)
    return _Conv2dFn.apply(x, w, stride, padding)
