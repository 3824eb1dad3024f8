"""Res2Net-101 backbone (scale=4, base width 26) — the [BJ:11] 512x512
milestone's encoder (hierarchical scale-split 3x3 convs
inside each bottleneck).  Random-init."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..blocks import ConvBNReLU, SodBatchNorm2d, SodConv2d, maxpool2d_3x3s2


class Bottle2neck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, planes: int, stride: int = 1,
                 scale: int = 4, base_width: int = 26):
        super().__init__()
        # width rounded UP to a multiple of 8 (26->32, 52->56): every split
        # then takes the vectorized/async-DMA kernel paths (the canonical
        # 26-wide splits ran the scalar BN/conv fallbacks — measured ~7% of
        # the 512^2 step).  MI355X-first design choice of this rebuild.
        width = (int(planes * base_width / 64.0) + 7) // 8 * 8
        self.scale = scale
        self.width = width
        self.stage = stride != 1 or cin != planes * self.expansion
        cout = planes * self.expansion
        self.conv1 = ConvBNReLU(cin, width * scale, 1)
        # scale-1 hierarchical 3x3 convs; the first sees no carry in a
        # "stage" block (standard Res2Net stype='stage' behavior)
        self.convs = nn.ModuleList([
            ConvBNReLU(width, width, 3, stride=stride, padding=1)
            for _ in range(scale - 1)
        ])
        self.stride = stride
        self.conv3 = SodConv2d(width * scale, cout, 1)
        self.bn3 = SodBatchNorm2d(cout)
        if self.stage:
            self.down_conv = SodConv2d(cin, cout, 1, stride=stride)
            self.down_bn = SodBatchNorm2d(cout)

    def forward(self, x):
        identity = x
        if self.stage:
            identity = self.down_bn(self.down_conv(x), relu=False)
        out = self.conv1(x)
        splits = torch.split(out, self.width, dim=1)
        ys = []
        prev = None
        for i in range(self.scale - 1):
            sp = splits[i] if (self.stage or prev is None) else splits[i] + prev
            prev = self.convs[i](sp)
            ys.append(prev)
        last = splits[self.scale - 1]
        if self.stage and self.stride > 1:
            last = F.avg_pool2d(last, 3, stride=self.stride, padding=1)
        ys.append(last)
        out = torch.cat(ys, dim=1)
        return self.bn3(self.conv3(out), residual=identity, relu=True)


class Res2NetBackbone(nn.Module):
    out_channels = [64, 256, 512, 1024, 2048]

    def __init__(self, layers=(3, 4, 23, 3), in_channels: int = 3):
        super().__init__()
        self.stem = ConvBNReLU(in_channels, 64, 7, stride=2, padding=3)
        cin = 64
        stages = []
        for planes, n, stride in zip([64, 128, 256, 512], layers, [1, 2, 2, 2]):
            blocks = [Bottle2neck(cin, planes, stride)]
            cin = planes * Bottle2neck.expansion
            blocks += [Bottle2neck(cin, planes) for _ in range(n - 1)]
            stages.append(nn.Sequential(*blocks))
        self.layer1, self.layer2, self.layer3, self.layer4 = stages

    def forward(self, x):
        c1 = self.stem(x)
        p = maxpool2d_3x3s2(c1)
        c2 = self.layer1(p)
        c3 = self.layer2(c2)
        c4 = self.layer3(c3)
        c5 = self.layer4(c4)
        return [c1, c2, c3, c4, c5]


def res2net101(in_channels: int = 3) -> Res2NetBackbone:
    return Res2NetBackbone((3, 4, 23, 3), in_channels)
