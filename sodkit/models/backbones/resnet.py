"""ResNet-50 feature extractor returning 5 stage features — parity with the
reference's backbone/ResNet.py `Backbone_ResNet50_in3`.
Random-init (BASELINE.json: no pretrained weights; datasets/checkpoints are
not fetchable).  Built on sodkit ops: the bottleneck's add+ReLU joins are
fused into the final BN (bn_act residual epilogue)."""

from __future__ import annotations

import torch.nn as nn

from ..blocks import ConvBNReLU, SodBatchNorm2d, SodConv2d, maxpool2d_3x3s2


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, planes: int, stride: int = 1):
        super().__init__()
        cout = planes * self.expansion
        self.conv1 = ConvBNReLU(cin, planes, 1)
        self.conv2 = ConvBNReLU(planes, planes, 3, stride=stride, padding=1)
        self.conv3 = SodConv2d(planes, cout, 1)
        self.bn3 = SodBatchNorm2d(cout)
        if stride != 1 or cin != cout:
            self.down_conv = SodConv2d(cin, cout, 1, stride=stride)
            self.down_bn = SodBatchNorm2d(cout)
        else:
            self.down_conv = None

    def forward(self, x):
        identity = x
        if self.down_conv is not None:
            identity = self.down_bn(self.down_conv(x), relu=False)
        y = self.conv2(self.conv1(x))
        # fused: BN(conv3) + identity, then ReLU — one bn_act call
        return self.bn3(self.conv3(y), residual=identity, relu=True)


class ResNetBackbone(nn.Module):
    """Returns [c1 (64, /2), c2 (256, /4), c3 (512, /8), c4 (1024, /16),
    c5 (2048, /32)] for a /1 input (SURVEY §2.3 stage table)."""

    out_channels = [64, 256, 512, 1024, 2048]

    def __init__(self, layers=(3, 4, 6, 3), in_channels: int = 3):
        super().__init__()
        self.stem = ConvBNReLU(in_channels, 64, 7, stride=2, padding=3)
        cin = 64
        stages = []
        for i, (planes, n, stride) in enumerate(
                zip([64, 128, 256, 512], layers, [1, 2, 2, 2])):
            blocks = [Bottleneck(cin, planes, stride)]
            cin = planes * Bottleneck.expansion
            blocks += [Bottleneck(cin, planes) for _ in range(n - 1)]
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


def resnet50(in_channels: int = 3) -> ResNetBackbone:
    return ResNetBackbone((3, 4, 6, 3), in_channels)
