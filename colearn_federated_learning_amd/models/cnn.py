"""The reference's own model family: a small MNIST-scale CNN.

CoLearn's runnable examples train a PySyft-tutorial-style two-conv CNN on
MNIST-shaped data; the [M] contract (§2b N13) names
logreg/ResNet/BERT, but a reference user switching over expects this
family too.  Convs and FCs route through the colearn op layer, so on
gfx950 the full fwd/bwd runs on the same HIP kernels as ResNet — no new
kernels, just a new composition.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from .base import CLinear, ColearnModel
from .resnet import _Conv2d


class MnistCNN(ColearnModel):
    """conv3x3(32)-relu-conv3x3(64)-relu-maxpool2-fc(128)-relu-fc(classes).

    Matches the classic PySyft/PyTorch MNIST example topology the
    reference repo's training scripts use.
    """

    def __init__(self, in_channels: int = 1, num_classes: int = 10,
                 image_size: int = 28) -> None:
        super().__init__()
        self.conv1 = _Conv2d(in_channels, 32, 3, padding=1)
        self.conv2 = _Conv2d(32, 64, 3, padding=1)
        feat = (image_size // 2) ** 2 * 64
        self.fc1 = CLinear(feat, 128)
        self.fc2 = CLinear(128, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = F.relu(self.conv1(x))
        x = F.relu(self.conv2(x))
        x = F.max_pool2d(x, 2)
        x = F.relu(self.fc1(x.flatten(1)))
        return self.fc2(x)
