"""BERT-base masked-LM on the colearn op layer.

Post-LN transformer encoder (BERT architecture: 12 layers, hidden 768,
12 heads, FFN 3072, GELU) with a tied-embedding MLM head.  All projections
route through ops.gemm (MFMA bf16 GEMM on gfx950), attention score/context
matmuls through ops.gemm.colearn_matmul, LayerNorms through ops.norms.
At the mandated seq_len=128 attention is two small batched GEMMs + a
softmax — no flash/long-context machinery is warranted (SURVEY.md §5,
long-context entry: explicitly out of scope).
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.gemm import colearn_linear, colearn_matmul, colearn_matmul_nt
from ..ops.norms import ColearnLayerNorm
from .base import ColearnModel


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden_size: int = 768
    num_layers: int = 12
    num_heads: int = 12
    intermediate_size: int = 3072
    max_position_embeddings: int = 512
    dropout: float = 0.1
    layer_norm_eps: float = 1e-12

    def __post_init__(self) -> None:
        if self.hidden_size % self.num_heads:
            raise ValueError(
                f"hidden_size {self.hidden_size} must be divisible by "
                f"num_heads {self.num_heads}")


class _Linear(nn.Module):
    def __init__(self, cin: int, cout: int) -> None:
        super().__init__()
        self.weight = nn.Parameter(torch.empty(cout, cin))
        self.bias = nn.Parameter(torch.zeros(cout))
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return colearn_linear(x, self.weight, self.bias)


class BertSelfAttention(nn.Module):
    def __init__(self, cfg: BertConfig) -> None:
        super().__init__()
        self.h = cfg.num_heads
        self.dh = cfg.hidden_size // cfg.num_heads
        self.qkv = _Linear(cfg.hidden_size, 3 * cfg.hidden_size)
        self.out = _Linear(cfg.hidden_size, cfg.hidden_size)
        self.drop = nn.Dropout(cfg.dropout)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b, s, d = x.shape
        qkv = self.qkv(x).view(b, s, 3, self.h, self.dh)
        q, k, v = (
            qkv[:, :, i].transpose(1, 2).reshape(b * self.h, s, self.dh)
            for i in range(3)
        )
        scores = colearn_matmul_nt(q, k) / math.sqrt(self.dh)
        # torch's softmax accumulates in fp32 internally for bf16
        probs = self.drop(F.softmax(scores, dim=-1))
        ctx = colearn_matmul(probs, v)
        ctx = ctx.view(b, self.h, s, self.dh).transpose(1, 2).reshape(b, s, d)
        return self.out(ctx)


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig) -> None:
        super().__init__()
        self.attn = BertSelfAttention(cfg)
        self.ln1 = ColearnLayerNorm(cfg.hidden_size, cfg.layer_norm_eps)
        self.fc1 = _Linear(cfg.hidden_size, cfg.intermediate_size)
        self.fc2 = _Linear(cfg.intermediate_size, cfg.hidden_size)
        self.ln2 = ColearnLayerNorm(cfg.hidden_size, cfg.layer_norm_eps)
        self.drop = nn.Dropout(cfg.dropout)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.ln1(x + self.drop(self.attn(x)))
        x = self.ln2(x + self.drop(self.fc2(F.gelu(self.fc1(x)))))
        return x


class BertForMaskedLM(ColearnModel):
    def __init__(self, cfg: BertConfig) -> None:
        super().__init__()
        self.cfg = cfg
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.pos_emb = nn.Embedding(cfg.max_position_embeddings,
                                    cfg.hidden_size)
        nn.init.normal_(self.tok_emb.weight, std=0.02)
        nn.init.normal_(self.pos_emb.weight, std=0.02)
        self.emb_ln = ColearnLayerNorm(cfg.hidden_size, cfg.layer_norm_eps)
        self.drop = nn.Dropout(cfg.dropout)
        self.layers = nn.ModuleList(
            BertLayer(cfg) for _ in range(cfg.num_layers)
        )
        # MLM head: transform + tied decoder (output embedding = tok_emb^T)
        self.head_fc = _Linear(cfg.hidden_size, cfg.hidden_size)
        self.head_ln = ColearnLayerNorm(cfg.hidden_size, cfg.layer_norm_eps)
        self.head_bias = nn.Parameter(torch.zeros(cfg.vocab_size))

    def encode(self, ids: torch.Tensor) -> torch.Tensor:
        b, s = ids.shape
        pos = torch.arange(s, device=ids.device)
        x = self.tok_emb(ids) + self.pos_emb(pos)[None]
        x = self.drop(self.emb_ln(x))
        for layer in self.layers:
            x = layer(x)
        return x

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        x = self.head_ln(F.gelu(self.head_fc(self.encode(ids))))
        return colearn_linear(x, self.tok_emb.weight, self.head_bias)

    def loss(self, x, y, params=None, buffers=None):
        """MLM loss with the head projected ONLY at (candidate) masked
        positions: the vocab projection is ~28% of full-step FLOPs but
        only ~15% of tokens are scored.  A fixed candidate budget (25% of
        tokens, >=17 sigma above the binomial mean) keeps every shape
        static, so the whole step stays hipGraph-capturable; candidates
        are selected with a stable argsort (graph-safe, unlike nonzero).
        Padding candidates carry label -100 and are ignored by the loss.
        """
        if params is not None:  # functional (DP) path: full composite
            merged = {**params, **(buffers or {})}
            logits = torch.func.functional_call(self, merged, (x,))
            return self._criterion(logits, y)
        hidden = self.encode(x)
        h = hidden.reshape(-1, hidden.shape[-1])
        flat_y = y.reshape(-1)
        k = max((flat_y.numel() * 25 + 99) // 100, 8)
        order = flat_y.ne(-100).to(torch.int8).argsort(
            descending=True, stable=True
        )[:k]
        hk = h.index_select(0, order)
        yk = flat_y.index_select(0, order)
        hk = self.head_ln(F.gelu(self.head_fc(hk)))
        logits = colearn_linear(hk, self.tok_emb.weight, self.head_bias)
        return F.cross_entropy(logits.float(), yk, ignore_index=-100)

    def _criterion(self, logits: torch.Tensor,
                   y: torch.Tensor) -> torch.Tensor:
        return F.cross_entropy(
            logits.float().view(-1, logits.shape[-1]), y.reshape(-1),
            ignore_index=-100,
        )

    @torch.no_grad()
    def loss_and_correct(
        self, x: torch.Tensor, y: torch.Tensor
    ) -> Tuple[torch.Tensor, int, int]:
        logits = self(x)
        loss = self._criterion(logits, y)
        mask = y != -100
        correct = int((logits.argmax(-1)[mask] == y[mask]).sum())
        return loss, correct, int(mask.sum())
