"""SOD quality metrics — parity with the reference's utils/metric.py
(MAE, precision/recall curve, max/mean F-measure with beta^2 = 0.3,
SURVEY §2.1).  numpy per-image implementations plus distributed-aware
accumulators (the eval all-reduce of SURVEY §3.2 / C5 lives in the meters)."""

from __future__ import annotations

import numpy as np
import torch
import torch.distributed as dist

BETA2 = 0.3
NBINS = 256


def cal_mae(pred: np.ndarray, mask: np.ndarray) -> float:
    """pred, mask float arrays in [0,1], same shape."""
    return float(np.abs(pred - mask).mean())


def cal_pr_maxf(pred: np.ndarray, mask: np.ndarray):
    """255-threshold precision/recall arrays + max-F for one image."""
    gt = mask > 0.5
    tp = np.zeros(NBINS)
    fp = np.zeros(NBINS)
    fn = np.zeros(NBINS)
    hist_fg, _ = np.histogram(pred[gt], bins=NBINS, range=(0, 1))
    hist_bg, _ = np.histogram(pred[~gt], bins=NBINS, range=(0, 1))
    # threshold t_k = k/NBINS: positives are bins >= k
    tp = np.cumsum(hist_fg[::-1])[::-1]
    fp = np.cumsum(hist_bg[::-1])[::-1]
    npos = gt.sum()
    fn = npos - tp
    prec = tp / np.maximum(tp + fp, 1)
    rec = tp / max(npos, 1)
    f = (1 + BETA2) * prec * rec / np.maximum(BETA2 * prec + rec, 1e-8)
    return prec, rec, float(f.max()), float(f.mean())


class MAEMeter:
    """Distributed MAE accumulator: per-image MAE summed locally, all-reduced
    once at compute()."""

    def __init__(self) -> None:
        self.total = 0.0
        self.count = 0

    @torch.no_grad()
    def update(self, pred: torch.Tensor, mask: torch.Tensor) -> None:
        # pred/mask [N,1,H,W] in [0,1]
        per_img = (pred.float() - mask.float()).abs().mean(dim=[1, 2, 3])
        self.total += float(per_img.sum().item())
        self.count += pred.shape[0]

    def compute(self, distributed: bool = False) -> float:
        total, count = self.total, self.count
        if distributed and dist.is_initialized() and dist.get_world_size() > 1:
            t = torch.tensor([total, float(count)], dtype=torch.float64)
            if torch.cuda.is_available():
                t = t.cuda()
            dist.all_reduce(t)
            total, count = t[0].item(), t[1].item()
        return total / max(count, 1)


class PRFMeter:
    """Distributed PR-curve / F-measure accumulator over NBINS thresholds.

    Reference protocol (utils/metric.py per SURVEY §2.1): precision/recall
    curves are computed PER IMAGE and macro-averaged across the dataset
    before the F-measure — NOT micro-averaged from pooled TP/FP histograms.
    The meter therefore sums per-image P/R arrays (linear, so the
    distributed all-reduce of the sums + image count is exact)."""

    def __init__(self) -> None:
        self.prec_sum = np.zeros(NBINS)
        self.rec_sum = np.zeros(NBINS)
        self.count = 0

    @torch.no_grad()
    def update(self, pred: torch.Tensor, mask: torch.Tensor) -> None:
        p = pred.float().clamp(0, 1)
        for i in range(p.shape[0]):
            pi = p[i].reshape(-1).cpu().numpy()
            mi = mask[i].float().reshape(-1).cpu().numpy()
            prec, rec, _, _ = cal_pr_maxf(pi, mi)
            self.prec_sum += prec
            self.rec_sum += rec
            self.count += 1

    def compute(self, distributed: bool = False):
        prec_sum, rec_sum, count = self.prec_sum, self.rec_sum, self.count
        if distributed and dist.is_initialized() and dist.get_world_size() > 1:
            t = torch.from_numpy(
                np.concatenate([prec_sum, rec_sum, [float(count)]]))
            if torch.cuda.is_available():
                t = t.cuda()
            dist.all_reduce(t)
            arr = t.cpu().numpy()
            prec_sum, rec_sum, count = arr[:NBINS], arr[NBINS:2 * NBINS], arr[-1]
        prec = prec_sum / max(count, 1)
        rec = rec_sum / max(count, 1)
        f = (1 + BETA2) * prec * rec / np.maximum(BETA2 * prec + rec, 1e-8)
        return {"precision": prec, "recall": rec,
                "maxf": float(f.max()), "meanf": float(f.mean())}
