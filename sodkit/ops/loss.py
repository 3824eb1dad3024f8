"""Fused hybrid pixel-wise loss: sigmoid + BCE + soft-IoU + SSIM — SURVEY
§2.3 K5, [BJ:5] "pixel-wise BCE+IoU/SSIM loss ... 11x11 gaussian window in
LDS ... fwd & analytic bwd".

GPU path (csrc/loss.hip):
  * ``loss_fwd``: one elementwise+reduce kernel (sigmoid, BCE sum, per-image
    IoU sums) + one windowed-stats kernel (11x11 gaussian SSIM, window row
    staged in LDS) producing the SSIM sum and the 5 stat maps saved for bwd.
  * ``loss_bwd``: analytic gradient — 3 coefficient center-maps from the
    saved stats, then one gather kernel (5 gaussian correlations) combined
    with the BCE/IoU terms and the sigmoid jacobian into dz.

The scalar composition (weights, per-image IoU mean) happens on [N]-sized
tensors in Python with no host sync.
"""

from __future__ import annotations

import torch

from . import reference
from ._backend import ext


class _HybridLossFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, z, t, w_bce, w_iou, w_ssim):
        z = z.contiguous(memory_format=torch.channels_last)
        t = t.to(torch.float32).contiguous(memory_format=torch.channels_last)
        bce_sum, img_sums, ssim_sum, p, stats = ext().loss_fwd(z, t, w_ssim > 0)
        ntot = float(z.numel())
        loss = z.new_zeros((), dtype=torch.float32)
        if w_bce:
            loss = loss + w_bce * bce_sum[0] / ntot
        if w_iou:
            inter = img_sums[:, 0]
            union = img_sums[:, 1] + img_sums[:, 2] - inter
            loss = loss + w_iou * (1.0 - (inter + 1.0) / (union + 1.0)).mean()
        if w_ssim:
            loss = loss + w_ssim * (1.0 - ssim_sum[0] / ntot)
        ctx.save_for_backward(p, t, img_sums, stats)
        ctx.meta = (w_bce, w_iou, w_ssim, z.dtype)
        return loss

    @staticmethod
    def backward(ctx, go):
        p, t, img_sums, stats = ctx.saved_tensors
        w_bce, w_iou, w_ssim, zdtype = ctx.meta
        # go stays a device tensor: no host sync in the step's backward
        go32 = go.reshape(1).to(device=p.device, dtype=torch.float32)
        dz = ext().loss_bwd(p, t, img_sums, stats, w_bce, w_iou, w_ssim, go32)
        return dz.to(zdtype), None, None, None, None


def hybrid_loss(logits: torch.Tensor, target: torch.Tensor,
                w_bce: float = 1.0, w_iou: float = 1.0,
                w_ssim: float = 1.0) -> torch.Tensor:
    if logits.is_cuda:
        return _HybridLossFn.apply(logits, target, w_bce, w_iou, w_ssim)
    return reference.hybrid_loss(logits, target, w_bce, w_iou, w_ssim)
