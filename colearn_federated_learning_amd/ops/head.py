"""Fused classifier head: avgpool + per-client FC + softmax cross-entropy.

One autograd op replaces the ~15-launch torch chain (adaptive_avg_pool2d,
batched matmul, float casts, softmax/nll, view/mean — fwd AND bwd) with 4
HIP launches (csrc/head.hip), part of killing the measured ~20%
small-kernel tail of the flagship step
(profiles/resnet18_fedavg_kernel_stats.md).

The FC weight/bias gradients are written straight into the flat-grad
slots (the params' ``.grad`` views) and the op returns None for them —
same direct-grad-write contract as the conv/BN kernels.
"""

from __future__ import annotations

import torch

from . import native_enabled, require_ext


class _FusedHeadLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, y, G, gw_slot, gb_slot):
        m = require_ext()
        loss_vec, pooled, probs = m.head_fwd(x, w, b, y, G)
        ctx.save_for_backward(pooled, probs, w, y, gw_slot, gb_slot)
        ctx.hw = (x.shape[2], x.shape[3])
        return loss_vec

    @staticmethod
    def backward(ctx, gloss):
        pooled, probs, w, y, gw_slot, gb_slot = ctx.saved_tensors
        m = require_ext()
        dx = m.head_bwd(probs, pooled, w, y,
                        gloss.contiguous().float(), gw_slot, gb_slot,
                        ctx.hw[0], ctx.hw[1])
        return dx, None, None, None, None, None, None


def fused_head_applicable(x: torch.Tensor, w: torch.Tensor,
                          num_classes: int) -> bool:
    # .grad views must exist: they are the direct wgrad targets (the
    # stacked engine's flat buffers attach them at construction)
    return (x.is_cuda and native_enabled() and x.dtype == torch.bfloat16
            and num_classes <= 32 and w.grad is not None
            and w.grad.is_contiguous())


def fused_head_loss(x: torch.Tensor, fc_w: torch.Tensor,
                    fc_b: torch.Tensor, y: torch.Tensor,
                    G: int) -> torch.Tensor:
    """Per-client mean CE loss [G] for channels-last bf16 features
    ``x`` [G*B, C, H, W] and per-client FC params [G, K, C] / [G, K]."""
    x = x.contiguous(memory_format=torch.channels_last)
    return _FusedHeadLoss.apply(x, fc_w, fc_b, y, G,
                                fc_w.grad, fc_b.grad)
