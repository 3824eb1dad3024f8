"""Hybrid pixel-wise SOD loss module — the reference's loss/ package
(CEL / HEL family) generalized to BCE + soft-IoU + SSIM per [BJ:5]."""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import hybrid_loss


class HybridLoss(nn.Module):
    def __init__(self, w_bce: float = 1.0, w_iou: float = 1.0,
                 w_ssim: float = 1.0):
        super().__init__()
        self.w_bce, self.w_iou, self.w_ssim = w_bce, w_iou, w_ssim

    def forward(self, logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        return hybrid_loss(logits, target, self.w_bce, self.w_iou, self.w_ssim)
