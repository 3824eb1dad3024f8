"""Bilinear resize with optional fused add — SURVEY §2.3 K4 (`cus_sample` /
`upsample_add` parity).  GPU: csrc/bilinear.hip (fwd gather + atomic-free
backward gather); CPU: F.interpolate."""

from __future__ import annotations

from typing import Optional

import torch

from . import reference
from ._backend import ext

_CL = torch.channels_last


def _cl(x: torch.Tensor) -> torch.Tensor:
    return x.contiguous(memory_format=_CL)


class _BilinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, out_h, out_w, add):
        x = _cl(x)
        y = ext().bilinear_fwd(x, out_h, out_w,
                               _cl(add) if add is not None else torch.Tensor())
        ctx.meta = (x.shape[2], x.shape[3], add is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        ih, iw, has_add = ctx.meta
        dy = _cl(dy)
        dx = ext().bilinear_bwd(dy, ih, iw)
        return dx, None, None, (dy if has_add else None)


def bilinear2d(x: torch.Tensor, out_h: int, out_w: int,
               add: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Resize x to (out_h, out_w), optionally fusing `+ add` (upsample_add)."""
    if x.shape[2] == out_h and x.shape[3] == out_w:
        return x if add is None else x + add
    if x.is_cuda:
        return _BilinearFn.apply(x, out_h, out_w, add)
    return reference.bilinear2d(x, out_h, out_w, add)


def upsample_add(low: torch.Tensor, high: torch.Tensor) -> torch.Tensor:
    return bilinear2d(low, high.shape[2], high.shape[3], add=high)
