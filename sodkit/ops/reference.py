"""Pure-PyTorch reference implementations of every sodkit op.

These serve two roles:
  1. the CPU execution path (the [BJ:7] plumbing config runs with no GPU);
  2. the fp32 numerics oracle the HIP kernels are unit-tested against
     (SURVEY §4.2.1).

They are differentiable torch code — autograd provides the CPU backward.
On a GPU device, sodkit ops never route here: the HIP extension is required
and fails loudly if missing (no silent eager fallback).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn.functional as F



def _fp(x: torch.Tensor) -> torch.Tensor:
    """Upcast to fp32 compute precision, but keep fp64 (gradcheck) intact."""
    return x if x.dtype == torch.float64 else x.float()

# ---------------------------------------------------------------------------
# Convolution.  sodkit stores conv weights in KRSC layout [Cout, R, S, Cin]
# (torch's channels-last weight layout): the MFMA implicit-GEMM A-operand is
# then W[cout][(r,s,cin)] row-major — coalesced LDS staging with no transpose.
# The reference path permutes to torch's [Cout, Cin, R, S].
# ---------------------------------------------------------------------------

def conv2d(x: torch.Tensor, w_krsc: torch.Tensor, bias: Optional[torch.Tensor],
           stride: int, padding: int, dilation: int) -> torch.Tensor:
    w = w_krsc.permute(0, 3, 1, 2)  # [Cout, Cin, R, S]
    return F.conv2d(x, w, bias, stride=stride, padding=padding, dilation=dilation)


# ---------------------------------------------------------------------------
# Fused BatchNorm(+residual add)(+ReLU).  Training-mode batch stats with
# optional cross-rank stats all-reduce (SyncBN).
# ---------------------------------------------------------------------------

class _SyncSum(torch.autograd.Function):
    """Autograd-correct cross-rank sum: forward all-reduces the stats,
    backward all-reduces the incoming gradient (the C4 backward-stats
    reduction real SyncBN needs — a transparent all_reduce under autograd
    would silently drop the cross-rank gradient terms)."""

    @staticmethod
    def forward(ctx, x, pg):
        ctx.pg = pg
        y = x.clone()
        torch.distributed.all_reduce(y, group=pg)
        return y

    @staticmethod
    def backward(ctx, g):
        g = g.contiguous().clone()
        torch.distributed.all_reduce(g, group=ctx.pg)
        return g, None


def bn_act(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
           running_mean: torch.Tensor, running_var: torch.Tensor,
           training: bool, momentum: float, eps: float,
           relu: bool = True, residual: Optional[torch.Tensor] = None,
           process_group=None) -> torch.Tensor:
    xf = _fp(x)
    if training:
        n = x.numel() // x.shape[1]
        dims = [0, 2, 3]
        s = xf.sum(dim=dims)
        ss = (xf * xf).sum(dim=dims)
        cnt = torch.full((1,), float(n), device=x.device)
        if process_group is not None and torch.distributed.is_initialized() \
                and torch.distributed.get_world_size(process_group) > 1:
            packed = _SyncSum.apply(torch.cat([s, ss, cnt]), process_group)
            s, ss, cnt = packed[: s.numel()], packed[s.numel():-1], packed[-1:]
        n_tot = cnt.item()
        mean = s / n_tot
        # clamp: the two-pass form can go slightly negative by cancellation
        var = (ss / n_tot - mean * mean).clamp_min(0.0)
        with torch.no_grad():
            unbiased = var * (n_tot / max(n_tot - 1.0, 1.0))
            running_mean.mul_(1 - momentum).add_(mean.detach(), alpha=momentum)
            running_var.mul_(1 - momentum).add_(unbiased.detach(), alpha=momentum)
    else:
        mean, var = running_mean, running_var
    invstd = torch.rsqrt(var.clamp_min(0.0) + eps)
    y = (xf - mean.view(1, -1, 1, 1)) * (invstd * gamma).view(1, -1, 1, 1) \
        + beta.view(1, -1, 1, 1)
    if residual is not None:
        y = y + _fp(residual)
    if relu:
        y = F.relu(y)
    return y.to(x.dtype)


# ---------------------------------------------------------------------------
# Bilinear resize (+ optional fused add) — `cus_sample` / `upsample_add`
# parity (SURVEY §2.1 utils/tensor_ops.py).
# ---------------------------------------------------------------------------

def bilinear2d(x: torch.Tensor, out_h: int, out_w: int,
               add: Optional[torch.Tensor] = None,
               align_corners: bool = False) -> torch.Tensor:
    y = F.interpolate(_fp(x), size=(out_h, out_w), mode="bilinear",
                      align_corners=align_corners)
    if add is not None:
        y = y + _fp(add)
    return y.to(x.dtype)


# ---------------------------------------------------------------------------
# Dynamic (spatially-variant) 3x3 filtering — HDFNet DDPM (K10).
# ---------------------------------------------------------------------------

def dynamic_filter(x: torch.Tensor, kernels: torch.Tensor,
                   dilation: int) -> torch.Tensor:
    """x [N,C,H,W], kernels [N,9,H,W] -> sum_e k_e * shift_e(x), zero pad."""
    n, c, h, w = x.shape
    cols = F.unfold(x.contiguous(), 3, dilation=dilation, padding=dilation)
    cols = cols.view(n, c, 9, h * w)
    k = kernels.reshape(n, 1, 9, h * w)
    return (cols * k).sum(dim=2).view(n, c, h, w)


# ---------------------------------------------------------------------------
# Hybrid pixel-wise loss: BCE-with-logits + soft IoU + SSIM (the [BJ:5]
# generalized loss; the reference's CEL/HEL family, SURVEY §2.1 loss/).
# All math in fp32.  Per-image IoU and SSIM, averaged over the batch.
# ---------------------------------------------------------------------------

_SSIM_WIN = 11
_SSIM_SIGMA = 1.5
_SSIM_C1 = 0.01 ** 2
_SSIM_C2 = 0.03 ** 2


def _gaussian_window(device, dtype=torch.float32):
    half = _SSIM_WIN // 2
    g = torch.tensor(
        [math.exp(-((i - half) ** 2) / (2 * _SSIM_SIGMA ** 2)) for i in range(_SSIM_WIN)],
        device=device, dtype=dtype)
    g = g / g.sum()
    w2d = torch.outer(g, g)
    return w2d.view(1, 1, _SSIM_WIN, _SSIM_WIN)


def ssim_map(p: torch.Tensor, t: torch.Tensor) -> torch.Tensor:
    """Per-pixel SSIM between probability map p and target t, both [N,1,H,W]
    fp32 in [0,1].  11x11 gaussian window, reflection-free (same padding via
    conv with zero pad is wrong at edges; use valid region weighting by
    renormalizing the window mass — implemented by dividing by the window sum
    of a ones image)."""
    w = _gaussian_window(p.device, p.dtype)
    pad = _SSIM_WIN // 2
    ones = torch.ones_like(p)
    wsum = F.conv2d(ones, w, padding=pad)            # window mass inside image
    mu_p = F.conv2d(p, w, padding=pad) / wsum
    mu_t = F.conv2d(t, w, padding=pad) / wsum
    s_pp = F.conv2d(p * p, w, padding=pad) / wsum - mu_p * mu_p
    s_tt = F.conv2d(t * t, w, padding=pad) / wsum - mu_t * mu_t
    s_pt = F.conv2d(p * t, w, padding=pad) / wsum - mu_p * mu_t
    num = (2 * mu_p * mu_t + _SSIM_C1) * (2 * s_pt + _SSIM_C2)
    den = (mu_p * mu_p + mu_t * mu_t + _SSIM_C1) * (s_pp + s_tt + _SSIM_C2)
    return num / den


def hybrid_loss(logits: torch.Tensor, target: torch.Tensor,
                w_bce: float = 1.0, w_iou: float = 1.0,
                w_ssim: float = 1.0) -> torch.Tensor:
    z = _fp(logits)
    t = _fp(target)
    loss = z.new_zeros(())
    if w_bce:
        loss = loss + w_bce * F.binary_cross_entropy_with_logits(z, t)
    p = torch.sigmoid(z)
    if w_iou:
        inter = (p * t).sum(dim=[1, 2, 3])
        union = p.sum(dim=[1, 2, 3]) + t.sum(dim=[1, 2, 3]) - inter
        iou = (inter + 1.0) / (union + 1.0)
        loss = loss + w_iou * (1.0 - iou).mean()
    if w_ssim:
        loss = loss + w_ssim * (1.0 - ssim_map(p, t).mean())
    return loss


# ---------------------------------------------------------------------------
# SGD with momentum + fp32 master weights (reference math for the fused
# multi-tensor kernel K11).  Operates in-place; grad scale folds in the
# 1/world_size gradient average and any loss-scale.
# ---------------------------------------------------------------------------

@torch.no_grad()
def sgd_step(param, grad, master32, momentum_buf32, lr: float, momentum: float,
             weight_decay: float, grad_scale: float, nesterov: bool = False):
    g = grad.float() * grad_scale
    if weight_decay:
        g = g.add(master32, alpha=weight_decay)
    momentum_buf32.mul_(momentum).add_(g)
    if nesterov:
        g = g.add(momentum_buf32, alpha=momentum)
    else:
        g = momentum_buf32
    master32.add_(g, alpha=-lr)
    param.copy_(master32.to(param.dtype))
