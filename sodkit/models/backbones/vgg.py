"""VGG-16 (BN) 5-stage feature extractor — parity with the reference's
backbone/VGG.py `Backbone_VGG16_in3`.  Random-init."""

from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ..blocks import ConvBNReLU


_CFG = [(64, 2), (128, 2), (256, 3), (512, 3), (512, 3)]


class VGGBackbone(nn.Module):
    """Returns [c1 (64, /1), c2 (128, /2), c3 (256, /4), c4 (512, /8),
    c5 (512, /16)]."""

    out_channels = [64, 128, 256, 512, 512]

    def __init__(self, in_channels: int = 3):
        super().__init__()
        stages = []
        cin = in_channels
        for cout, n in _CFG:
            convs = []
            for _ in range(n):
                convs.append(ConvBNReLU(cin, cout, 3, padding=1))
                cin = cout
            stages.append(nn.Sequential(*convs))
        self.stages = nn.ModuleList(stages)

    def forward(self, x):
        feats = []
        for i, stage in enumerate(self.stages):
            if i > 0:
                x = F.max_pool2d(x, 2, stride=2)
            x = stage(x)
            feats.append(x)
        return feats


def vgg16(in_channels: int = 3) -> VGGBackbone:
    return VGGBackbone(in_channels)
