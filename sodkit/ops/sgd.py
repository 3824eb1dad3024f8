"""SodSGD — SGD-with-momentum + fp32 master weights, fused multi-tensor HIP
kernel on GPU (one launch sweeps every parameter chunk;
the bf16->fp32 grad upcast, 1/world grad average, loss-scale unscale, weight
decay, momentum and the fp32->bf16 param write-back all fold into it).

"sgd_trick" parity (reference config): no weight decay on bias / BN params —
callers pass two param groups via ``split_decay_groups``.

CPU path: per-param reference math (ops.reference.sgd_step).
"""

from __future__ import annotations

from typing import Iterable

import torch

from . import reference
from ._backend import ext

_CHUNK = 16384  # elements per kernel block-chunk (must match csrc/sgd.hip)


def split_decay_groups(model: torch.nn.Module):
    """Return (decay, no_decay) parameter lists: biases and BN gamma/beta
    (ndim <= 1) get no weight decay."""
    decay, no_decay = [], []
    for name, p in model.named_parameters():
        if not p.requires_grad:
            continue
        (no_decay if p.ndim <= 1 else decay).append(p)
    return decay, no_decay


class SodSGD:
    """Not a torch.optim.Optimizer subclass on purpose: the fused step needs
    a per-step (lr, grad_scale) and owns its flat table layout."""

    def __init__(self, param_groups, momentum: float = 0.9,
                 weight_decay: float = 5e-4, nesterov: bool = False):
        # param_groups: list of dicts {params: [...], weight_decay: float}
        self.groups = []
        for g in param_groups:
            params = [p for p in g["params"] if p.requires_grad]
            self.groups.append({"params": params,
                                "weight_decay": float(g.get("weight_decay", weight_decay))})
        self.momentum = momentum
        self.nesterov = nesterov
        self.state: dict[torch.nn.Parameter, dict] = {}
        for g in self.groups:
            for p in g["params"]:
                self.state[p] = {
                    "master": p.detach().to(torch.float32).clone(),
                    "momentum_buffer": torch.zeros_like(p, dtype=torch.float32),
                }
        self._tables = None  # lazily built device pointer tables (GPU)
        # deferred-grad mode (reducer.defer): .grad is a fresh autograd
        # tensor each step, so the trainer points grad_src at the reducer's
        # STABLE bucket views — the gather in finalize() lands grads there
        # and the baked pointer tables stay valid.  In view mode the views
        # ARE .grad, so wiring it unconditionally is a no-op.
        self.grad_src: "dict[torch.nn.Parameter, torch.Tensor] | None" = None

    # -- fused GPU tables (one table + one launch per param dtype: conv
    #    weights are bf16/fp16 while BN affine params stay fp32) -----------
    def _build_tables(self):
        groups_by_dtype: dict[torch.dtype, list] = {}
        for g in self.groups:
            wd = g["weight_decay"]
            for p in g["params"]:
                g = (self.grad_src.get(p) if self.grad_src is not None
                     else None)
                if g is None:
                    g = p.grad
                assert g is not None, "grads must be pre-assigned (reducer bucket views)"
                st = self.state[p]
                groups_by_dtype.setdefault(p.dtype, []).append(
                    ((p.data_ptr(), g.data_ptr(), st["master"].data_ptr(),
                      st["momentum_buffer"].data_ptr()), p.numel(), wd, p.device))
        self._tables = []
        codes = {torch.float32: 0, torch.bfloat16: 1, torch.float16: 2}
        for dtype, entries in groups_by_dtype.items():
            dev = entries[0][3]
            ptr_t = torch.tensor([e[0] for e in entries], dtype=torch.int64, device=dev)
            numel_t = torch.tensor([e[1] for e in entries], dtype=torch.int64, device=dev)
            wd_t = torch.tensor([e[2] for e in entries], dtype=torch.float32, device=dev)
            chunks = []
            for ti, e in enumerate(entries):
                for off in range(0, e[1], _CHUNK):
                    chunks.append((ti, off))
            chunk_t = torch.tensor(chunks, dtype=torch.int64, device=dev)
            self._tables.append((codes[dtype], ptr_t, numel_t, wd_t, chunk_t))

    @torch.no_grad()
    def step(self, lr, grad_scale: float = 1.0,
             scale_t: "torch.Tensor | None" = None):
        """``scale_t`` (GPU only): a device fp32 scalar multiplied into
        grad_scale INSIDE the kernel; 0.0 skips the update entirely.  This is
        the sync-free fp16 dynamic-scaler contract — the overflow decision
        never touches the host.  ``lr`` may be a device fp32 scalar tensor
        (hipGraph capture: the host updates the tensor between replays)."""
        first = next((p for g in self.groups for p in g["params"]), None)
        if first is None:
            return
        if first.is_cuda:
            if self._tables is None:
                self._build_tables()
            st = scale_t if scale_t is not None else torch.Tensor()
            if isinstance(lr, torch.Tensor):
                lr_t, lr_f = lr, 0.0
            else:
                lr_t, lr_f = torch.Tensor(), float(lr)
            for dt, ptr_t, numel_t, wd_t, chunk_t in self._tables:
                ext().sgd_fused(ptr_t, numel_t, wd_t, chunk_t, dt, lr_f,
                                self.momentum, float(grad_scale),
                                self.nesterov, st, lr_t)
            # weights just changed: re-transpose every cached dgrad w_t in
            # one launch (sodkit/ops/conv.py _WtCache)
            from . import conv as _conv
            _conv._WtCache.refresh()
        else:
            lr = float(lr)
            for g in self.groups:
                for p in g["params"]:
                    if p.grad is None:
                        continue
                    st = self.state[p]
                    reference.sgd_step(p.data, p.grad, st["master"],
                                       st["momentum_buffer"], lr, self.momentum,
                                       g["weight_decay"], grad_scale, self.nesterov)

    # -- checkpointing (bitwise resume of optimizer state) -------------------
    def state_dict(self) -> dict:
        flat = []
        for g in self.groups:
            for p in g["params"]:
                st = self.state[p]
                flat.append({"master": st["master"],
                             "momentum_buffer": st["momentum_buffer"]})
        return {"momentum": self.momentum, "nesterov": self.nesterov,
                "param_state": flat,
                "group_wd": [g["weight_decay"] for g in self.groups]}

    def load_state_dict(self, sd: dict) -> None:
        self.momentum = sd["momentum"]
        self.nesterov = sd["nesterov"]
        it = iter(sd["param_state"])
        for g in self.groups:
            for p in g["params"]:
                st = next(it)
                self.state[p]["master"].copy_(st["master"])
                self.state[p]["momentum_buffer"].copy_(st["momentum_buffer"])
        for g, wd in zip(self.groups, sd["group_wd"]):
            g["weight_decay"] = wd
        self._tables = None

    def params(self) -> Iterable[torch.nn.Parameter]:
        for g in self.groups:
            yield from g["params"]
