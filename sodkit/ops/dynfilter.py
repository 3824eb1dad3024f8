"""Dynamic (spatially-variant) 3x3 filtering — the HDFNet DDPM op (SURVEY
§2.3 K10).  GPU: dedicated gather kernels (csrc/dynfilter.hip) — no unfold
materialization; CPU (and shape corner cases): the composed reference."""

from __future__ import annotations

import torch

from . import reference
from ._backend import ext

_CL = torch.channels_last


def _fast_ok(x: torch.Tensor) -> bool:
    c8 = x.shape[1] // 8
    return (x.is_cuda and x.shape[1] % 8 == 0 and c8 <= 64
            and (c8 & (c8 - 1)) == 0 and x.dtype in (torch.bfloat16, torch.float16))


class _DynFilterFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, dilation):
        x = x.contiguous(memory_format=_CL)
        k = k.to(x.dtype).contiguous(memory_format=_CL)
        y = ext().dynfilter_fwd(x, k, dilation)
        ctx.save_for_backward(x, k)
        ctx.dilation = dilation
        return y

    @staticmethod
    def backward(ctx, dy):
        x, k = ctx.saved_tensors
        dx, dk = ext().dynfilter_bwd(
            dy.contiguous(memory_format=_CL), x, k, ctx.dilation)
        return dx, dk, None


def dynamic_filter(x: torch.Tensor, kernels: torch.Tensor,
                   dilation: int) -> torch.Tensor:
    if _fast_ok(x):
        return _DynFilterFn.apply(x, kernels, dilation)
    return reference.dynamic_filter(x, kernels, dilation)
