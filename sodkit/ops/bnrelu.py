"""Fused BatchNorm(+residual add)(+ReLU) — SURVEY §2.3 K3 + K8 epilogue.

GPU path: hand-written HIP kernels (csrc/bnrelu.hip):
  * ``bn_stats``      one-pass per-channel sum / sumsq reduction (fp32)
  * ``bn_apply``      normalize + scale + (add residual) + (ReLU), one pass
  * ``bn_bwd_reduce`` per-channel sum(g), sum(g*xhat) with g = dy·relu-mask
  * ``bn_bwd_apply``  dx (+ optional residual grad) in one pass

SyncBN is the same op with a process group: the per-channel
stats (and backward reduction terms) are all-reduced once per call as a single
coalesced [2, C] buffer (per-rank counts are equal by construction —
DistributedSampler drop_last — so no count exchange and no host sync), and
the kernel path is identical to the non-sync one.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from . import reference
from ._backend import ext

_CL = torch.channels_last

# single-launch cooperative BN forward (csrc/bn_fused.hip) — NEGATIVE
# RESULT, opt-in only (SODKIT_BN_FUSED=1).  The hypothesis was that
# collapsing stats+finalize+apply into one grid.sync'd kernel kills two
# ~5us launch floors per layer; measured: 810 vs 1213 imgs/s — the
# cooperative launch + two grid-wide syncs per layer cost far more than
# the floors they save (the co-resident grid is also smaller than the
# tuned per-kernel grids).  Numerics-identical and fully tested; kept
# as documentation.
import os
_BNF_ENV = os.environ.get("SODKIT_BN_FUSED", "")


def _cl(x: torch.Tensor) -> torch.Tensor:
    return x.contiguous(memory_format=_CL)


class _BNActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training,
                momentum, eps, relu, residual, process_group, slabs):
        x = _cl(x)
        if residual is not None:
            residual = _cl(residual)
        C = x.shape[1]
        n_local = x.numel() // C
        fused = False
        if training:
            sync = (process_group is not None and dist.is_initialized()
                    and dist.get_world_size(process_group) > 1)
            if sync:
                # conv-emitted slabs skip the stats re-read of x here too
                stats = (ext().bn_slab_sum(slabs) if slabs is not None
                         else ext().bn_stats(x))           # fp32 [2, C]
                dist.all_reduce(stats, group=process_group)
                # equal per-rank shapes by construction (DistributedSampler
                # with drop_last) -> no count exchange, no host sync
                n_total = float(n_local) * dist.get_world_size(process_group)
                mean, invstd = ext().bn_finalize(stats, n_total, momentum,
                                                 eps, running_mean,
                                                 running_var, True)
            elif slabs is not None:
                # conv->BN stats fusion (conv2d_fwd_stats): finalize the
                # conv's per-tile (sum, sumsq) partials — no y re-read
                n_total = float(n_local)
                mean, invstd = ext().bn_finalize_slabs(slabs, n_total,
                                                       momentum, eps,
                                                       running_mean,
                                                       running_var)
            elif _BNF_ENV == "1" and ext().bn_fwd_fused_eligible(
                    C, x.element_size()):
                n_total = float(n_local)
                y, mask, mean, invstd = ext().bn_fwd_fused(
                    x, gamma, beta, n_total, momentum, eps, running_mean,
                    running_var, relu,
                    residual if residual is not None else torch.Tensor(),
                    True)
                fused = True
            else:
                n_total = float(n_local)
                mean, invstd = ext().bn_stats_finalize(x, n_total, momentum,
                                                       eps, running_mean,
                                                       running_var)
        else:
            n_total = 0.0                                   # eval marker
            mean, invstd = ext().bn_finalize(running_mean, 0.0, momentum, eps,
                                             running_mean, running_var, False)
        if not fused:
            # training: also emit the 1-bit ReLU-mask plane so the backward
            # kernels read 1/16 of y's bytes for the mask instead of y itself
            y, mask = ext().bn_apply(
                x, mean, invstd, gamma, beta, relu,
                residual if residual is not None else torch.Tensor(), training)
        ctx.save_for_backward(x, y, mean, invstd, gamma,
                              mask if mask.numel() else torch.Tensor())
        ctx.meta = (relu, residual is not None, n_total, process_group)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, mean, invstd, gamma, mask = ctx.saved_tensors
        relu, has_res, n_total, pg = ctx.meta
        dy = _cl(dy)
        sums = ext().bn_bwd_reduce(dy, y, x, mean, invstd, relu,
                                   mask)  # fp32 [2, C]
        if pg is not None and dist.is_initialized() and dist.get_world_size(pg) > 1:
            dist.all_reduce(sums, group=pg)
        dx, dres = ext().bn_bwd_apply(dy, y, x, mean, invstd, gamma, sums,
                                      n_total, relu, has_res, mask)
        dgamma = sums[1].to(gamma.dtype)
        dbeta = sums[0].to(gamma.dtype)
        return (dx, dgamma, dbeta, None, None, None, None, None, None,
                dres if has_res else None, None, None)


def bn_act(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
           running_mean: torch.Tensor, running_var: torch.Tensor,
           training: bool, momentum: float = 0.1, eps: float = 1e-5,
           relu: bool = True, residual: Optional[torch.Tensor] = None,
           process_group=None) -> torch.Tensor:
    if x.is_cuda:
        # conv-emitted stat slabs (sodkit.ops.conv._apply_conv) ride the
        # producing tensor; consume-and-remove so they can never go stale
        slabs = getattr(x, "_sodkit_stats", None)
        if slabs is not None:
            del x._sodkit_stats
            if not (training and slabs.dim() == 2
                    and slabs.shape[1] == 2 * x.shape[1]):
                slabs = None
        return _BNActFn.apply(x, gamma, beta, running_mean, running_var,
                              training, momentum, eps, relu, residual,
                              process_group, slabs)
    return reference.bn_act(x, gamma, beta, running_mean, running_var,
                            training, momentum, eps, relu, residual,
                            process_group)
