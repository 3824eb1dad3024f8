"""The reference's model-specific losses (SURVEY §2.1 loss/):

* CEL — consistency-enhanced loss (MINet, CVPR'20):
      L = (FP + FN) / (TP + FP + FN)
        = (sum p + sum t - 2 sum pt) / (sum p + sum t - sum pt), per image.
* HEL — hybrid enhanced loss (HDFNet, ECCV'20): edge-weighted + region terms;
  reconstructed as boundary-band-weighted L1 on the probability map plus the
  foreground/background consistency ratio.

Both are cheap elementwise+reduction torch compositions (autograd provides
backward on CPU and GPU); the headline BASELINE config uses the fused
BCE+IoU+SSIM kernel (losses.hybrid) — these exist for loss-surface parity
and are selectable via Config.loss.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class CELLoss(nn.Module):
    """BCE + CEL (the MINet recipe)."""

    def __init__(self, w_bce: float = 1.0, w_cel: float = 1.0):
        super().__init__()
        self.w_bce = w_bce
        self.w_cel = w_cel

    def forward(self, logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        z = logits.float()
        t = target.float()
        loss = z.new_zeros(())
        if self.w_bce:
            loss = loss + self.w_bce * F.binary_cross_entropy_with_logits(z, t)
        p = torch.sigmoid(z)
        inter = (p * t).sum(dim=[1, 2, 3])
        sp = p.sum(dim=[1, 2, 3])
        st = t.sum(dim=[1, 2, 3])
        cel = (sp + st - 2 * inter) / (sp + st - inter + 1e-6)
        return loss + self.w_cel * cel.mean()


class HELLoss(nn.Module):
    """Hybrid enhanced loss: edge-band weighted term + region consistency
    (HDFNet-style reconstruction)."""

    def __init__(self, edge_ks: int = 5, w_edge: float = 1.0,
                 w_region: float = 1.0):
        super().__init__()
        self.edge_ks = edge_ks
        self.w_edge = w_edge
        self.w_region = w_region

    def forward(self, logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        z = logits.float()
        t = target.float()
        p = torch.sigmoid(z)
        # edge band: dilation(t) - erosion(t) via maxpool
        k, pad = self.edge_ks, self.edge_ks // 2
        dil = F.max_pool2d(t, k, stride=1, padding=pad)
        ero = -F.max_pool2d(-t, k, stride=1, padding=pad)
        edge = dil - ero
        edge_term = (edge * (p - t).abs()).sum(dim=[1, 2, 3]) \
            / (edge.sum(dim=[1, 2, 3]) + 1e-6)
        fg = ((1 - p) * t).sum(dim=[1, 2, 3]) / (t.sum(dim=[1, 2, 3]) + 1e-6)
        bg = (p * (1 - t)).sum(dim=[1, 2, 3]) \
            / ((1 - t).sum(dim=[1, 2, 3]) + 1e-6)
        return (self.w_edge * edge_term + self.w_region * (fg + bg)).mean()


def build_loss(cfg):
    """Config.loss -> criterion module (default: the fused hybrid kernel)."""
    from .hybrid import HybridLoss
    name = getattr(cfg, "loss", "hybrid")
    if name == "hybrid":
        return HybridLoss(cfg.w_bce, cfg.w_iou, cfg.w_ssim)
    if name == "cel":
        return CELLoss(cfg.w_bce)
    if name == "hel":
        return HELLoss()
    raise KeyError(f"unknown loss {name!r}")
