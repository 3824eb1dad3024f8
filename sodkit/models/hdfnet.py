"""HDFNet — RGB-D SOD with hierarchical dynamic filtering (capability parity
with the reference's network/HDFNet.py: two-stream encoder (RGB + depth),
depth-to-RGB dense transport layers, and a dynamic-dilated-pyramid decoder
generating spatially-variant kernels from depth features — SURVEY §2.1,
[BJ:10]).  Reconstruction (reference mount empty).

DDPM: depth-derived features predict a per-pixel 3x3 kernel per dilation
d in {1, 2, 4}; each is applied to the RGB decoder feature as a dynamic
(spatially-variant) filter; the pyramid sum passes through Conv+BN+ReLU.
Dynamic filtering runs
on the dedicated K10 HIP kernels (csrc/dynfilter.hip) on GPU.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import bilinear2d, dynamic_filter, upsample_add
from .blocks import ConvBNReLU, SodConv2d
from .backbones import resnet50, vgg16


class DenseTransLayer(nn.Module):
    """Depth→RGB transport: depth feature modulated by an RGB/depth joint
    gate, producing the kernel-source feature for DDPM."""

    def __init__(self, ch: int = 64):
        super().__init__()
        self.depth_conv = ConvBNReLU(ch, ch, 3, padding=1)
        self.gate_conv = SodConv2d(2 * ch, ch, 1, bias=True)

    def forward(self, rgb: torch.Tensor, depth: torch.Tensor) -> torch.Tensor:
        gate = torch.sigmoid(self.gate_conv(torch.cat([rgb, depth], dim=1)))
        return self.depth_conv(depth) * gate


class DDPM(nn.Module):
    """Dynamic Dilated Pyramid Module."""

    dilations = (1, 2, 4)

    def __init__(self, ch: int = 64):
        super().__init__()
        self.kgen = nn.ModuleList([SodConv2d(ch, 9, 3, padding=1, bias=True)
                                   for _ in self.dilations])
        self.fuse = ConvBNReLU(ch, ch, 3, padding=1)

    def forward(self, feat: torch.Tensor, kfeat: torch.Tensor) -> torch.Tensor:
        y = None
        for kg, d in zip(self.kgen, self.dilations):
            k = torch.softmax(kg(kfeat), dim=1)        # normalized kernels
            yd = dynamic_filter(feat, k, d)
            y = yd if y is None else y + yd
        return self.fuse(y)


class HDFNet(nn.Module):
    def __init__(self, backbone: str = "res50", ch: int = 64):
        super().__init__()
        if backbone == "res50":
            self.rgb_backbone = resnet50(3)
            self.depth_backbone = resnet50(1)
        else:
            self.rgb_backbone = vgg16(3)
            self.depth_backbone = vgg16(1)
        bc = self.rgb_backbone.out_channels
        self.rgb_trans = nn.ModuleList([ConvBNReLU(c, ch, 1) for c in bc])
        self.dep_trans = nn.ModuleList([ConvBNReLU(c, ch, 1) for c in bc[2:]])
        self.translayers = nn.ModuleList([DenseTransLayer(ch) for _ in range(3)])
        self.ddpms = nn.ModuleList([DDPM(ch) for _ in range(3)])
        self.up_convs = nn.ModuleList([ConvBNReLU(ch, ch, 3, padding=1)
                                       for _ in range(4)])
        self.head = ConvBNReLU(ch, 32, 3, padding=1)
        self.classifier = SodConv2d(32, 1, 1, bias=True)

    def forward(self, rgb: torch.Tensor, depth: torch.Tensor) -> torch.Tensor:
        H, W = rgb.shape[2], rgb.shape[3]
        rf = self.rgb_backbone(rgb)
        df = self.depth_backbone(depth)
        t = [tr(f) for tr, f in zip(self.rgb_trans, rf)]
        # depth transport on the three deepest levels (c3, c4, c5)
        kf = [self.translayers[i](t[i + 2], self.dep_trans[i](df[i + 2]))
              for i in range(3)]
        d = self.ddpms[2](t[4], kf[2])                       # level 5
        d = self.ddpms[1](self.up_convs[3](upsample_add(d, t[3])), kf[1])
        d = self.ddpms[0](self.up_convs[2](upsample_add(d, t[2])), kf[0])
        d = self.up_convs[1](upsample_add(d, t[1]))
        d = self.up_convs[0](upsample_add(d, t[0]))
        logits = self.classifier(self.head(d))
        return bilinear2d(logits, H, W)
