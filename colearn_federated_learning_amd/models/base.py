"""Model base class: forward + loss in one object.

The reference's models are plain ``nn.Module`` subclasses trained with an
external criterion.  Here the loss lives on the
model so the round engine is model-agnostic (classification and MLM models
plug into the same client/coordinator code), and so the DP path can take
functional per-sample gradients via ``torch.func.functional_call``.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


class ColearnModel(nn.Module):
    """Forward produces logits; subclasses may override loss for MLM etc."""

    def loss(
        self,
        x: torch.Tensor,
        y: torch.Tensor,
        params: Optional[Dict[str, torch.Tensor]] = None,
        buffers: Optional[Dict[str, torch.Tensor]] = None,
    ) -> torch.Tensor:
        if params is not None:
            merged = {**params, **(buffers or {})}
            logits = torch.func.functional_call(self, merged, (x,))
        else:
            logits = self(x)
        return self._criterion(logits, y)

    def _criterion(self, logits: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        return F.cross_entropy(logits.float(), y)

    @torch.no_grad()
    def loss_and_correct(
        self, x: torch.Tensor, y: torch.Tensor
    ) -> Tuple[torch.Tensor, int, int]:
        """Returns (loss, correct_predictions, predictions_scored)."""
        logits = self(x)
        loss = self._criterion(logits, y)
        correct = int((logits.argmax(dim=-1) == y).sum())
        return loss, correct, y.numel()


class CLinear(nn.Module):
    """nn.Linear-compatible module routed through the colearn GEMM (so the
    module CALL is visible to hooks — the per-sample grad engine needs it)."""

    def __init__(self, in_features: int, out_features: int) -> None:
        super().__init__()
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.empty(out_features))
        # identical init to nn.Linear
        nn.init.kaiming_uniform_(self.weight, a=5 ** 0.5)
        bound = 1 / (in_features ** 0.5)
        nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops.gemm import colearn_linear

        return colearn_linear(x, self.weight, self.bias)


class LogisticRegression(ColearnModel):
    """Multinomial logistic regression — the plumbing-config model."""

    def __init__(self, in_features: int, num_classes: int) -> None:
        super().__init__()
        self.fc = CLinear(in_features, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fc(x.flatten(1))
