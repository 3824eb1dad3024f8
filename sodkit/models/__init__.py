"""Model registry — `MINet_VGG16` / `MINet_Res50` / `HDFNet_Res50` parity
 plus the [BJ:11] Res2Net101 variant."""

from __future__ import annotations

import torch.nn as nn

from .blocks import (ConvBNReLU, SodBatchNorm2d, SodConv2d,  # noqa: F401
                     convert_precision, set_syncbn)
from .minet import MINet
from .hdfnet import HDFNet
from . import backbones


def build_model(name: str) -> nn.Module:
    name = name.lower()
    if name == "minet_res50":
        return MINet(backbones.resnet50(3))
    if name == "minet_vgg16":
        return MINet(backbones.vgg16(3))
    if name == "minet_res2net101":
        return MINet(backbones.res2net101(3))
    if name == "hdfnet_res50":
        return HDFNet("res50")
    if name == "hdfnet_vgg16":
        return HDFNet("vgg16")
    raise KeyError(f"unknown model {name!r}")


MODEL_NAMES = ["minet_res50", "minet_vgg16", "minet_res2net101",
               "hdfnet_res50", "hdfnet_vgg16"]
