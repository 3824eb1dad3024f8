"""Convolution op: MFMA implicit-GEMM HIP kernels on GPU, ATen on CPU.

Weight layout is KRSC ``[Cout, R, S, Cin]`` (torch's channels-last weight
layout): W[cout][(r,s,cin)] row-major IS the implicit-GEMM A-operand, staged
into LDS with coalesced loads and no transpose.  Activations are
NHWC-physical (torch ``channels_last``) on GPU, plain NCHW on CPU.

Kernels (csrc/conv_igemm.hip): K1/K2/K6/K10/K13 of SURVEY §2.3 — one
implicit-GEMM family covering 1x1 (pure GEMM), 3x3, 7x7 stem, strided and
dilated cases; dgrad via transposed-weight implicit GEMM; wgrad via split-K
NHW-reduction GEMM.
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from . import reference
from ._backend import ext

_CL = torch.channels_last

# 8-phase 256-square conv backend (csrc/conv_igemm8.hip, docs/
# IGEMM8_PLAN.md).  Measured A/B across the model shape table
# (profiles/r2/ubench_conv_final.log): the 256x256 tile only beats the 128-square
# kernel when the launch fills the chip — its win shapes have
# (ceil(M/256) * Cout/256) >= ~256 workgroups (e.g. 1x1 expand 64->256
# @88x88: 143 vs 114 TF); below that it underfills 256 CUs and LOSES
# (3x3 512ch @11x11: 110 vs 323 TF).  Dispatch is therefore gated on the
# workgroup count; SODKIT_IGEMM_8P=1/0 forces it on/off for A/B runs.
# Threshold from the full r2 table (profiles/r2/ubench_conv_final.log): every
# measured shape at >= 244 wgs wins (l3_1x1e fwd +30%, l3_1x1r dgrad
# +23%, l2_1x1e, l1_1x1e), every shape at <= 128 loses; 232 splits them.
_8P_ENV = os.environ.get("SODKIT_IGEMM_8P", "")
_8P_MIN_WG = 232


def _use_8p(m_tiles: int, cout: int) -> bool:
    if _8P_ENV == "0":
        return False
    if _8P_ENV == "1":
        return True
    return m_tiles * (cout // 256) >= _8P_MIN_WG


# direct sliding-window wgrad for 3x3 s1 d1 p1 (csrc/wgrad_dir.hip): stages
# each x row once and computes all nine taps from shifted LDS reads instead
# of re-gathering per tap.  SODKIT_WGRAD_DIR=0 disables (A/B).
# Measured (r2 ubench wdir column): 426 vs 166 TF at W=176, 251-253 vs
# 160 at W=88, a wash at W=44, and a LOSS at W<=22 (k-tile padding +
# short rows) — so the gate is W >= 64.
_WDIR_ENV = os.environ.get("SODKIT_WGRAD_DIR", "")


def _use_wgrad_dir(w: torch.Tensor, x: torch.Tensor, stride, padding,
                   dilation) -> bool:
    if _WDIR_ENV == "0":
        return False
    return (w.shape[1] == 3 and w.shape[2] == 3 and stride == 1
            and padding == 1 and dilation == 1 and w.shape[3] % 64 == 0
            and 64 <= x.shape[3] <= 184)


# VALU 1x1 wgrad (csrc/wgrad_k1.hip) — NEGATIVE RESULT, opt-in only
# (SODKIT_WGRAD_K1=1).  Hypothesis was that the MFMA path's transpose
# staging (108-187 TF) left room for a register-accumulating VALU
# reduction; measured r2: 10-38 TF — the per-pixel convert+FMA issue
# stream is ~5x slower than the staged MFMA math even though it avoids
# the transpose.  Kept (numerics-tested) as documentation; do not
# re-enable without restructuring around packed dot2 and wider m-tiles.
_WK1_ENV = os.environ.get("SODKIT_WGRAD_K1", "")


def _use_wgrad_k1(w: torch.Tensor, padding, dilation) -> bool:
    if _WK1_ENV != "1":
        return False
    return (w.shape[1] == 1 and w.shape[2] == 1 and padding == 0
            and dilation == 1 and w.shape[3] % 16 == 0
            and w.shape[0] % 4 == 0 and w.shape[0] >= 4)


# direct fwd/dgrad for the 64-channel 3x3 decoder/AIM/SIM class
# (csrc/conv_dir.hip): the im2col kernel re-stages every x element 9x.
# Measured (r2): W=176 fwd/dgrad 517/522 vs igemm 436/400; W=88 went
# from a loss (293 vs 364) to a WIN (377/378 vs 366/356) once the
# ROWS=2 path pooled the row-pair's q-fragments across all wave slots,
# so the gate is W >= 64.  SODKIT_CONV_DIR=0 disables (A/B).
_CDIR_ENV = os.environ.get("SODKIT_CONV_DIR", "")


def _use_conv_dir(cin: int, cout: int, w3: bool, x: torch.Tensor, stride,
                  padding, dilation) -> bool:
    if _CDIR_ENV == "0":
        return False
    return (w3 and stride == 1 and padding == 1 and dilation == 1
            and cin in (32, 64) and cout <= 64 and cout % 4 == 0
            and 64 <= x.shape[3] <= 184)


def _cl_ld(x: torch.Tensor):
    """Return (tensor, channel pitch).  A channel-slice of a channels_last
    tensor (torch.split / cat-backward narrow views) keeps NHWC pixel layout
    with pitch = parent C; the kernels take that pitch directly, so the
    slice costs no copy."""
    if x.is_contiguous(memory_format=_CL) or (x.shape[1] == 1 and x.is_contiguous()):
        return x, x.shape[1]
    st = x.stride()
    if (x.dim() == 4 and st[1] == 1 and st[2] == x.shape[3] * st[3]
            and st[0] == x.shape[2] * st[2] and st[3] >= x.shape[1]):
        return x, st[3]
    x = x.contiguous(memory_format=_CL)
    return x, x.shape[1]


def _cl(x: torch.Tensor) -> torch.Tensor:
    return x.contiguous(memory_format=_CL)


class _ConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, stride, padding, dilation, emit_stats):
        x, ldx = _cl_ld(x)
        b = bias if bias is not None else torch.Tensor()
        if w.shape[0] % 256 == 0 and w.shape[3] % 8 == 0:
            p = (x.shape[2] + 2 * padding - dilation * (w.shape[1] - 1) - 1) \
                // stride + 1
            q = (x.shape[3] + 2 * padding - dilation * (w.shape[2] - 1) - 1) \
                // stride + 1
            m_tiles = -(-x.shape[0] * p * q // 256)
            use8 = _use_8p(m_tiles, w.shape[0])
        else:
            use8 = False
        slabs = None
        if use8:
            if emit_stats:
                y, slabs = ext().conv2d_fwd_8p_stats(x, w, b, stride, padding,
                                                     dilation, ldx)
            else:
                y = ext().conv2d_fwd_8p(x, w, b, stride, padding, dilation,
                                        ldx)
        elif _use_conv_dir(w.shape[3], w.shape[0],
                           w.shape[1] == 3 and w.shape[2] == 3,
                           x, stride, padding, dilation):
            if emit_stats:
                y, slabs = ext().conv2d_fwd_dir_stats(x, w, b, ldx)
            else:
                y = ext().conv2d_fwd_dir(x, w, b, ldx)
        elif emit_stats:
            # ES epilogue: the kernel also writes per-channel BN stat
            # partials [tiles_m, 2*Cout] so a following BatchNorm skips its
            # stats re-read of y entirely (docs/KERNELS.md)
            y, slabs = ext().conv2d_fwd_stats(x, w, b, stride, padding,
                                              dilation, ldx)
        else:
            y = ext().conv2d_fwd(x, w, b, stride, padding, dilation, ldx)
        if slabs is None:
            slabs = torch.empty(0, device=x.device, dtype=torch.float32)
        ctx.save_for_backward(x, w)
        ctx.meta = (stride, padding, dilation, bias is not None, ldx)
        ctx.mark_non_differentiable(slabs)
        # without this, autograd MATERIALIZES a zero-filled grad for the
        # slab output on every backward (~110 FillFunctor launches/step,
        # ~0.7 ms — seen in profiles/r2/kernel_summary_final.txt)
        ctx.set_materialize_grads(False)
        return y, slabs

    @staticmethod
    def backward(ctx, dy, _dslabs):
        if dy is None:   # materialize_grads(False): y unused in the graph
            return None, None, None, None, None, None, None
        x, w = ctx.saved_tensors
        stride, padding, dilation, has_bias, ldx = ctx.meta
        dy, lddy = _cl_ld(dy)
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            # dgrad A-operand layout [Cin, R, S, Cout] (cached; see _WtCache)
            w_t = _WtCache.get(w)
            if w_t.shape[0] % 256 == 0 and w_t.shape[3] % 8 == 0 and _use_8p(
                    -(-x.shape[0] * x.shape[2] * x.shape[3] // 256),
                    w_t.shape[0]):
                dx = ext().conv2d_dgrad_8p(dy, w_t, x.shape[2], x.shape[3],
                                           stride, padding, dilation, lddy)
            elif _use_conv_dir(w_t.shape[3], w_t.shape[0],
                               w_t.shape[1] == 3 and w_t.shape[2] == 3,
                               x, stride, padding, dilation):
                # dgrad roles: GEMM channels = Cout(=64), output = Cin
                dx = ext().conv2d_dgrad_dir(dy, w_t, lddy)
            else:
                dx = ext().conv2d_dgrad(dy, w_t, x.shape[2], x.shape[3],
                                        stride, padding, dilation, lddy)
        if ctx.needs_input_grad[1]:
            if _use_wgrad_dir(w, x, stride, padding, dilation):
                dw = ext().conv2d_wgrad_dir(x, dy, ldx, lddy).to(w.dtype)
            elif _use_wgrad_k1(w, padding, dilation):
                dw = ext().conv2d_wgrad_k1(x, dy, stride, ldx,
                                           lddy).to(w.dtype)
            else:
                dw = ext().conv2d_wgrad(x, dy, w.shape[1], w.shape[2],
                                        stride, padding, dilation, ldx,
                                        lddy).to(w.dtype)
        if has_bias and ctx.needs_input_grad[2]:
            db = ext().channel_sum(dy, lddy).to(w.dtype)
        return dx, dw, db, None, None, None, None


# conv->BN stats fusion (SODKIT_CONV_STATS=0 disables): the slabs ride the
# output tensor as `._sodkit_stats`; sodkit.ops.bn_act consumes (and
# removes) them when the very next op is a BatchNorm over that tensor.
_CST_ENV = os.environ.get("SODKIT_CONV_STATS", "")

# dgrad w_t cache (SODKIT_WT_CACHE=0 disables): materializing
# w.permute(3,1,2,0).contiguous() per conv per step was ~110 ATen copy
# launches (~0.64 ms/step, profiles/r2/kernel_summary_final.txt).  Weights
# only change inside the fused SGD, which calls refresh() — ONE
# multipermute kernel re-transposes every cached buffer in place.  Any
# OTHER mutation is caught: a torch-op write bumps `_version`, a `.data`
# replacement changes `data_ptr` — both invalidate the entry and the
# backward falls back to the eager permute (and re-registers).
_WT_ENV = os.environ.get("SODKIT_WT_CACHE", "")


class _WtCache:
    token = 0
    entries: dict = {}   # data_ptr -> entry dict
    tables = None
    dirty = False

    @classmethod
    def get(cls, w: torch.Tensor) -> torch.Tensor:
        if _WT_ENV == "0" or w.element_size() != 2:
            return w.permute(3, 1, 2, 0).contiguous()
        key = w.data_ptr()
        ent = cls.entries.get(key)
        if (ent is not None and ent["version"] == w._version
                and ent["token"] == cls.token):
            return ent["buf"]
        buf = w.permute(3, 1, 2, 0).contiguous()
        cls.entries[key] = {
            "version": w._version, "token": cls.token, "buf": buf,
            # keep the source storage alive — this is load-bearing TWICE:
            # (1) a pending refresh after a `.data` swap must not read
            # freed memory, and (2) pinning the storage means its data_ptr
            # can never be recycled to a NEW same-shape weight, so the
            # data_ptr KEY uniquely identifies this weight forever (a
            # recycled pointer + fresh _version could otherwise alias a
            # stale entry and serve another model's transposed weights).
            # Cost: a dead model's cached weights stay resident until the
            # process exits — noise against 288 GB of HBM.
            "src_t": w.detach(),
            "K": w.shape[0], "RS": w.shape[1] * w.shape[2], "C": w.shape[3],
        }
        cls.dirty = True
        return buf

    @classmethod
    def refresh(cls) -> None:
        """Called by SodSGD.step right after the weight update: bump the
        epoch token and re-transpose every cached w_t in one launch."""
        cls.token += 1
        if not cls.entries:
            return
        if cls.dirty or cls.tables is None:
            if torch.cuda.is_current_stream_capturing():
                return   # stale entries -> eager fallback inside capture
            ents = list(cls.entries.values())
            dev = ents[0]["buf"].device

            def mk(vals, dt):
                return torch.tensor(vals, dtype=dt, device=dev)

            seg, rsb, ktb, ctb = [], [], [], []
            for i, e in enumerate(ents):
                for rs in range(e["RS"]):
                    for kt in range((e["K"] + 63) // 64):
                        for ct in range((e["C"] + 63) // 64):
                            seg.append(i)
                            rsb.append(rs)
                            ktb.append(kt)
                            ctb.append(ct)
            cls.tables = (
                mk([e["src_t"].data_ptr() for e in ents], torch.int64),
                mk([e["buf"].data_ptr() for e in ents], torch.int64),
                mk([e["K"] for e in ents], torch.int32),
                mk([e["C"] for e in ents], torch.int32),
                mk([e["RS"] for e in ents], torch.int32),
                mk(seg, torch.int32), mk(rsb, torch.int32),
                mk(ktb, torch.int32), mk(ctb, torch.int32),
            )
            cls.dirty = False
        ext().multipermute(*cls.tables)
        for e in cls.entries.values():
            e["token"] = cls.token


def _apply_conv(x, w, bias, stride, padding, dilation):
    emit = (_CST_ENV != "0" and torch.is_grad_enabled()
            and (x.requires_grad or w.requires_grad))
    y, slabs = _ConvFn.apply(x, w, bias, stride, padding, dilation, emit)
    if slabs.numel():
        y._sodkit_stats = slabs
    return y


def conv2d(x: torch.Tensor, w_krsc: torch.Tensor,
           bias: Optional[torch.Tensor] = None, stride: int = 1,
           padding: int = 0, dilation: int = 1) -> torch.Tensor:
    if x.is_cuda:
        cin = x.shape[1]
        if cin < 8 and cin % 8 != 0 and not x.requires_grad:
            # stem-style inputs (RGB=3 / depth=1; no input grad): the old
            # pad-to-8-channels route wasted 5/8 of the gather and MFMA
            # (stem measured 41 TF fwd / 39 TF wgrad).  Materialize the
            # im2col matrix once instead — the stem becomes a 1x1 conv over
            # K8 channels and fwd/wgrad/bias ride the plain GEMM paths
            # (dgrad is never needed here).  dw flows back through the
            # differentiable reshape/pad of w.
            R, S = w_krsc.shape[1], w_krsc.shape[2]
            K = R * S * cin
            K8 = (K + 7) // 8 * 8
            xc, ldx = _cl_ld(x)
            xim = ext().im2col_nhwc(xc, R, S, stride, padding, dilation, ldx)
            w1 = w_krsc.reshape(w_krsc.shape[0], K)
            if K8 != K:
                import torch.nn.functional as F
                w1 = F.pad(w1, (0, K8 - K))
            w1 = w1.reshape(w_krsc.shape[0], 1, 1, K8)
            return _apply_conv(xim, w1, bias, 1, 0, 1)
        return _apply_conv(x, w_krsc, bias, stride, padding, dilation)
    return reference.conv2d(x, w_krsc, bias, stride, padding, dilation)
