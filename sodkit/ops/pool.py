"""Stem max pool — SURVEY §2.3 K7 (reference: ResNet/VGG stem pooling).

GPU path (csrc/maxpool.hip): NHWC 3x3 stride-2 pad-1 max pool whose
forward records a packed 4-bit argmax per channel; the backward is a
deterministic per-input-pixel GATHER over the <= 4 covering windows (no
atomics).  ATen's nhwc pair measured 86/190 us on the stem tensor; this
pair is bounded by one read + one write of each tensor.  Other pool
geometries (VGG 2x2) stay on ATen — they are not in any hot stem.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from ._backend import ext

_CL = torch.channels_last


class _MaxPool3x3s2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y, idx = ext().maxpool3x3s2_fwd(x)
        ctx.save_for_backward(idx)
        ctx.hw = (x.shape[2], x.shape[3])
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        H, W = ctx.hw
        return ext().maxpool3x3s2_bwd(dy, idx, H, W)


def maxpool2d_3x3s2(x: torch.Tensor) -> torch.Tensor:
    if (x.is_cuda and x.element_size() == 2 and x.shape[1] % 8 == 0
            and x.is_contiguous(memory_format=_CL)):
        return _MaxPool3x3s2Fn.apply(x)
    return F.max_pool2d(x, 3, stride=2, padding=1)
