"""Fused multi-head self-attention (BERT shapes: seq <= 128, dh = 64).

One forward + two backward HIP launches per (batch of heads) replace the
~12-kernel torch chain per attention layer (score bmm, scale, fp32
softmax, dropout, context bmm, reshape copies — fwd and bwd).  Dropout
is philox-keyed on (seed, head-block, element): deterministic,
layout-independent, and regenerated in the backward (csrc/attn.hip).
"""

from __future__ import annotations

import torch

from . import native_enabled, require_ext


class _FusedAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, heads, scale, p, seed):
        out, P, mask = require_ext().attn_fwd(qkv, heads, scale, p, seed)
        ctx.save_for_backward(qkv, P, mask)
        ctx.args = (heads, scale, p, seed)
        return out

    @staticmethod
    def backward(ctx, dctx):
        qkv, P, mask = ctx.saved_tensors
        heads, scale, p, seed = ctx.args
        dqkv = require_ext().attn_bwd(dctx.contiguous(), qkv, P, mask,
                                      heads, scale, p, seed)
        return dqkv, None, None, None, None


def fused_attention_applicable(qkv: torch.Tensor) -> bool:
    """qkv [N, S, 3, H, dh]: native path needs bf16 CUDA, dh 64 and
    S == 128 exactly — the kernel's fragment layout is specialized to
    the BERT-base MLM config (8 key-fragments per wave); other shapes
    fall back to the batched-GEMM composition."""
    return (qkv.is_cuda and native_enabled()
            and qkv.dtype == torch.bfloat16 and qkv.shape[-1] == 64
            and qkv.shape[1] == 128)


def fused_attention(qkv: torch.Tensor, heads: int, scale: float,
                    p: float, seed: int) -> torch.Tensor:
    """Returns ctx [N, S, H*dh]."""
    return _FusedAttnFn.apply(qkv.contiguous(), heads, scale, p, seed)
