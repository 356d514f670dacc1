"""ResNet-18 / ResNet-50 on the colearn op layer.

Standard He-initialized ResNet v1 with the CIFAR-style small-input stem
(3x3 stride-1, no maxpool) when ``small_input``, or the ImageNet stem
otherwise.  All convs route through ops.conv (implicit-GEMM HIP on gfx950),
norms through ops.norms (fused HIP BatchNorm), the classifier through
ops.gemm — the full fwd/bwd hot path is ours.  ``group_norm=True`` swaps
BatchNorm for GroupNorm(32) — required in DP mode, where per-sample
gradients are incompatible with cross-sample batch statistics.
"""

from __future__ import annotations

from typing import List, Optional, Type

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.conv import colearn_conv2d
from ..ops.norms import ColearnBatchNorm2d
from .base import CLinear, ColearnModel


class _Conv2d(nn.Module):
    """Conv with K-MAJOR weight storage [CO, KH, KW, CI] — the HIP conv
    kernels' native operand layout (zero weight permutes in the hot path).
    Init draws in the standard [CO, CI, KH, KW] layout first so the
    distribution (and RNG stream consumption) matches nn.Conv2d."""

    def __init__(self, cin: int, cout: int, k: int, stride: int = 1,
                 padding: int = 0) -> None:
        super().__init__()
        w = torch.empty(cout, cin, k, k)
        nn.init.kaiming_normal_(w, mode="fan_out", nonlinearity="relu")
        self.weight = nn.Parameter(w.permute(0, 2, 3, 1).contiguous())
        self.stride, self.padding = stride, padding

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return colearn_conv2d(x, self.weight, self.stride, self.padding)


class _GNAct(nn.Module):
    """GroupNorm with the same (x, residual) interface as the fused BN —
    the DP-mode substitute (eager add + relu)."""

    def __init__(self, planes: int, relu: bool) -> None:
        super().__init__()
        self.gn = nn.GroupNorm(min(32, planes), planes)
        self.do_relu = relu

    def forward(self, x: torch.Tensor, residual=None) -> torch.Tensor:
        y = self.gn(x)
        if residual is not None:
            y = y + residual
        return F.relu(y) if self.do_relu else y


def _norm(planes: int, group_norm: bool, relu: bool = False) -> nn.Module:
    if group_norm:
        return _GNAct(planes, relu)
    # relu/residual fuse into the BN kernel sweeps on gfx950
    return ColearnBatchNorm2d(planes, relu=relu)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes: int, planes: int, stride: int,
                 group_norm: bool) -> None:
        super().__init__()
        self.conv1 = _Conv2d(inplanes, planes, 3, stride, 1)
        self.bn1 = _norm(planes, group_norm, relu=True)
        self.conv2 = _Conv2d(planes, planes, 3, 1, 1)
        self.bn2 = _norm(planes, group_norm, relu=True)  # fused +idt, relu
        self.downsample: Optional[nn.Module] = None
        if stride != 1 or inplanes != planes:
            self.downsample = nn.Sequential(
                _Conv2d(inplanes, planes, 1, stride, 0),
                _norm(planes, group_norm),
            )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        idt = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), residual=idt)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes: int, planes: int, stride: int,
                 group_norm: bool) -> None:
        super().__init__()
        width = planes
        self.conv1 = _Conv2d(inplanes, width, 1, 1, 0)
        self.bn1 = _norm(width, group_norm, relu=True)
        self.conv2 = _Conv2d(width, width, 3, stride, 1)
        self.bn2 = _norm(width, group_norm, relu=True)
        self.conv3 = _Conv2d(width, planes * 4, 1, 1, 0)
        self.bn3 = _norm(planes * 4, group_norm, relu=True)  # fused +idt
        self.downsample: Optional[nn.Module] = None
        if stride != 1 or inplanes != planes * 4:
            self.downsample = nn.Sequential(
                _Conv2d(inplanes, planes * 4, 1, stride, 0),
                _norm(planes * 4, group_norm),
            )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        idt = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=idt)


class ResNet(ColearnModel):
    def __init__(
        self,
        block: Type[nn.Module],
        layers: List[int],
        num_classes: int = 10,
        in_channels: int = 3,
        small_input: bool = True,
        group_norm: bool = False,
    ) -> None:
        super().__init__()
        self.inplanes = 64
        if small_input:
            self.stem = nn.Sequential(
                _Conv2d(in_channels, 64, 3, 1, 1),
                _norm(64, group_norm, relu=True),
            )
            self.pool = nn.Identity()
        else:
            self.stem = nn.Sequential(
                _Conv2d(in_channels, 64, 7, 2, 3),
                _norm(64, group_norm, relu=True),
            )
            self.pool = nn.MaxPool2d(3, 2, 1)
        self.layer1 = self._make_layer(block, 64, layers[0], 1, group_norm)
        self.layer2 = self._make_layer(block, 128, layers[1], 2, group_norm)
        self.layer3 = self._make_layer(block, 256, layers[2], 2, group_norm)
        self.layer4 = self._make_layer(block, 512, layers[3], 2, group_norm)
        self.fc = CLinear(512 * block.expansion, num_classes)

    def _make_layer(self, block, planes: int, n: int, stride: int,
                    group_norm: bool) -> nn.Sequential:
        blocks = [block(self.inplanes, planes, stride, group_norm)]
        self.inplanes = planes * block.expansion
        for _ in range(1, n):
            blocks.append(block(self.inplanes, planes, 1, group_norm))
        return nn.Sequential(*blocks)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.pool(self.stem(x))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def resnet18(**kw) -> ResNet:
    return ResNet(BasicBlock, [2, 2, 2, 2], **kw)


def resnet50(**kw) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], **kw)
