"""Building-block modules — parity with the reference's module/BaseBlocks.py
(`BasicConv2d` = Conv+BN+ReLU) — built on sodkit.ops.

Weights live in KRSC layout [Cout, R, S, Cin] (torch's channels-last weight
layout): the MFMA implicit-GEMM A-operand is then W[cout][(r,s,cin)]
row-major, so the MI355X kernels stage it into LDS coalesced with no
transpose.  BN affine params and running stats stay fp32 regardless of the
compute dtype; `convert_precision` casts only conv weights/biases.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from .. import ops


class SodConv2d(nn.Module):
    def __init__(self, cin: int, cout: int, k: int, stride: int = 1,
                 padding: int = 0, dilation: int = 1, bias: bool = False):
        super().__init__()
        self.cin, self.cout, self.k = cin, cout, k
        self.stride, self.padding, self.dilation = stride, padding, dilation
        self.weight = nn.Parameter(torch.empty(cout, k, k, cin))
        # kaiming-normal fan_out (torchvision resnet convention)
        fan_out = k * k * cout
        with torch.no_grad():
            self.weight.normal_(0, math.sqrt(2.0 / fan_out))
        if bias:
            self.bias = nn.Parameter(torch.zeros(cout))
        else:
            self.register_parameter("bias", None)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.conv2d(x, self.weight, self.bias, self.stride,
                          self.padding, self.dilation)

    def extra_repr(self) -> str:  # pragma: no cover
        return (f"{self.cin}->{self.cout} k{self.k} s{self.stride} "
                f"p{self.padding} d{self.dilation}")


class SodBatchNorm2d(nn.Module):
    """BatchNorm2d with fused (residual-add +) ReLU epilogue and optional
    SyncBN (one coalesced stats all-reduce per call).
    The forward signature carries `residual`/`relu` so the fusion happens at
    the op, not as separate kernels."""

    def __init__(self, c: int, momentum: float = 0.1, eps: float = 1e-5):
        super().__init__()
        self.c = c
        self.momentum = momentum
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(c))    # gamma (fp32 always)
        self.bias = nn.Parameter(torch.zeros(c))     # beta
        self.register_buffer("running_mean", torch.zeros(c))
        self.register_buffer("running_var", torch.ones(c))
        self.process_group = None                    # set by set_syncbn

    def forward(self, x: torch.Tensor, residual: Optional[torch.Tensor] = None,
                relu: bool = True) -> torch.Tensor:
        return ops.bn_act(x, self.weight, self.bias, self.running_mean,
                          self.running_var, self.training, self.momentum,
                          self.eps, relu, residual, self.process_group)


class ConvBNReLU(nn.Module):
    """`BasicConv2d` parity: Conv + BN + ReLU (THE fusion target [BJ:5])."""

    def __init__(self, cin: int, cout: int, k: int, stride: int = 1,
                 padding: int = 0, dilation: int = 1, relu: bool = True):
        super().__init__()
        self.conv = SodConv2d(cin, cout, k, stride, padding, dilation)
        self.bn = SodBatchNorm2d(cout)
        self.relu = relu

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        return self.bn(self.conv(x), residual=residual, relu=self.relu)


def maxpool2d_3x3s2(x: torch.Tensor) -> torch.Tensor:
    """Stem maxpool: custom NHWC argmax-gather pair on GPU
    (sodkit/ops/pool.py — ATen's nhwc backward ran at ~0.3 TB/s)."""
    return ops.maxpool2d_3x3s2(x)


def convert_precision(model: nn.Module, dtype: torch.dtype) -> nn.Module:
    """Cast conv weights/biases to the compute dtype; BN stays fp32."""
    for m in model.modules():
        if isinstance(m, SodConv2d):
            m.weight.data = m.weight.data.to(dtype)
            if m.bias is not None:
                m.bias.data = m.bias.data.to(dtype)
    return model


def set_syncbn(model: nn.Module, process_group=None) -> nn.Module:
    """Enable SyncBN.  `process_group=None` selects the default (WORLD)
    group; the op treats `process_group is None` on the module as sync-off,
    so the sentinel WORLD object is stored explicitly."""
    import torch.distributed as dist
    if process_group is None:
        process_group = dist.group.WORLD
    for m in model.modules():
        if isinstance(m, SodBatchNorm2d):
            m.process_group = process_group
    return model
