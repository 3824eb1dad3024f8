"""Solver — the trainer/solver loop: epoch loop,
per-iter poly LR, (loss-scaled) fp16/bf16 step, custom gradient engine,
checkpoint save/resume, distributed eval with MAE/F-measure.

One step = load -> fwd -> hybrid loss -> bwd (buckets all-reduce on the side
stream as they fill) -> fused SGD.  The 1/world average and any loss-scale
unscale fold into the SGD grad_scale.
"""

from __future__ import annotations

import os
import time
from typing import Optional

import torch
import torch.distributed as dist

from .. import ops
from ..config import Config
from ..data import create_loader
from ..losses import build_loss
from ..metrics import MAEMeter, PRFMeter
from ..models import build_model, convert_precision, set_syncbn
from ..parallel import GradReducer, setup, get_rank, get_world_size
from ..utils import AvgMeter, get_logger, seed_everything
from ..utils.png import write_gray_png
from ..utils.recorder import ResultsRecorder, TBRecorder
from .checkpoint import exp_paths, load_checkpoint, make_exp_tree, save_checkpoint

_DTYPES = {"bf16": torch.bfloat16, "fp16": torch.float16, "fp32": torch.float32}


class Solver:
    def __init__(self, cfg: Config):
        self.cfg = cfg
        self.device = setup(cfg.backend, cfg.device)
        self.rank = get_rank()
        self.world = get_world_size()
        self.distributed = self.world > 1
        self.dtype = _DTYPES[cfg.dtype]
        self.is_gpu = self.device.type == "cuda"

        self.paths = exp_paths(cfg.output_root, cfg.exp_name)
        if self.rank == 0:
            make_exp_tree(self.paths)
            with open(self.paths["cfg"], "w") as f:
                f.write(cfg.to_json())
        self.log = get_logger("sodkit", self.paths["log"] if self.rank == 0 else None)

        # identical seed on every rank -> identical random init
        seed_everything(cfg.seed)
        model = build_model(cfg.model)
        if self.is_gpu and self.dtype == torch.float32:
            # the MFMA conv kernels are bf16/fp16 (fp32 accumulate); a full
            # f32-MFMA conv path (mfma_f32_16x16x4f32) is not built — fail
            # here with guidance instead of deep in the first conv dispatch
            raise ValueError(
                "dtype=fp32 on GPU is not supported: the CDNA4 conv kernels "
                "compute in bf16/fp16 with fp32 accumulation (BN/loss/SGD "
                "already run fp32 internally). Use --dtype bf16 (default) "
                "or fp16; fp32 end-to-end runs on CPU (device=cpu).")
        if self.is_gpu:
            # NOTE: no model.to(memory_format=channels_last) — sodkit weights
            # are KRSC-contiguous by construction and activations are made
            # channels_last at the input / by the kernels themselves.
            model = model.to(self.device)
            if self.dtype != torch.float32:
                convert_precision(model, self.dtype)
        if cfg.sync_bn and self.distributed:
            set_syncbn(model, None)  # default process group
        self.model = model

        self.reducer = GradReducer(model, bucket_mb=cfg.bucket_mb,
                                   overlap=cfg.overlap_comm)
        self.reducer.broadcast_params(model)

        decay, no_decay = ops.split_decay_groups(model)
        groups = [{"params": decay, "weight_decay": cfg.weight_decay}]
        if cfg.sgd_trick:
            groups.append({"params": no_decay, "weight_decay": 0.0})
        else:
            groups[0]["params"] = decay + no_decay
        self.optimizer = ops.SodSGD(groups, momentum=cfg.momentum,
                                    weight_decay=cfg.weight_decay,
                                    nesterov=cfg.nesterov)
        # stable grad source for the fused-SGD pointer tables (deferred-grad
        # mode re-lands autograd's per-step grad tensors in these views)
        self.optimizer.grad_src = self.reducer.grad_views

        self.criterion = build_loss(cfg)
        self.loss_scale = cfg.loss_scale if cfg.dtype == "fp16" else 1.0
        # simple dynamic loss scaling for fp16 (bf16 needs none): on a
        # non-finite grad norm the step is skipped and the scale halves;
        # after `growth_interval` good steps it doubles.
        self.dynamic_scale = cfg.dtype == "fp16"
        self._good_steps = 0
        self._growth_interval = 500
        self.skipped_steps = 0  # host-path overflow skips (visibility)
        # GPU: the whole scaler/clip decision lives in device tensors and the
        # fused SGD consumes a device multiplier (0.0 == skip) — the fp16
        # step path has NO per-step host sync.  iter_count still advances on
        # (device-decided) skipped steps: the host cannot know without a
        # sync, and a handful of early-run skips shift the poly schedule
        # negligibly.  Skips are counted device-side and logged.
        self._use_device_scaler = self.is_gpu and (self.dynamic_scale
                                                   or cfg.grad_clip > 0)
        if self._use_device_scaler:
            self._scale_t = torch.tensor(self.loss_scale, device=self.device)
            self._good_t = torch.zeros((), device=self.device)
            self._skips_t = torch.zeros((), device=self.device)

        self.start_epoch = 0
        self.iter_count = 0
        self.total_iters = 1
        self._accum_count = 0
        self.tb = None

        # hipGraph whole-step capture (cfg.hip_graph): single-GPU,
        # accum_steps == 1 only — the RCCL collectives stay out of capture
        # until graph+RCCL is validated on a multi-GPU node.
        self._graph_ok = (cfg.hip_graph and self.is_gpu
                          and not self.distributed
                          and max(1, cfg.accum_steps) == 1)
        self._graph = None
        self._graph_warm = 0
        self._g_static = None
        self._g_ev = None      # input-copy retirement fence (one step ahead)
        self._g_hold = None    # keeps the in-flight batch's buffers alive
        self._h2d_pending = []  # CPU(pinned) source tensors of enqueued H2Ds

    # ------------------------------------------------------------------
    def _lr_at(self, it: int) -> float:
        cfg = self.cfg
        if cfg.warmup_iters and it < cfg.warmup_iters:
            return cfg.lr * (it + 1) / cfg.warmup_iters
        if cfg.lr_type == "poly":
            frac = min(it, self.total_iters - 1) / max(self.total_iters, 1)
            return cfg.lr * (1.0 - frac) ** cfg.lr_power
        return cfg.lr

    def _to_device(self, t: torch.Tensor, cast: bool) -> torch.Tensor:
        if self._graph_ok and not t.is_cuda:
            # graph mode: the replay loop's host side runs far ahead of the
            # device, so the loader would recycle this pinned buffer before
            # the async H2D executes — hold it until the fence retires it
            self._h2d_pending.append(t)
        t = t.to(self.device, non_blocking=True)
        if cast and self.is_gpu and self.dtype != torch.float32:
            t = t.to(self.dtype)
        if self.is_gpu and self.cfg.channels_last and t.ndim == 4:
            t = t.contiguous(memory_format=torch.channels_last)
        return t

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _device_clip_and_scale(self) -> torch.Tensor:
        """GPU path: grad norm, overflow detection, clipping and the dynamic
        loss-scale state update computed entirely in device tensors.  Returns
        the fp32 device scalar the fused SGD multiplies into grad_scale
        (0.0 => the kernel skips the update).  No ``.item()`` — the fp16
        [BJ:11] config runs sync-free."""
        cfg = self.cfg
        sq = None
        for b in self.reducer.buckets:
            n = b.flat.float().square().sum()
            sq = n if sq is None else sq + n
        accum = max(1, cfg.accum_steps)
        # true (averaged, unscaled) grad norm; _scale_t is 1.0 unless fp16
        norm = torch.sqrt(sq) * (self.reducer.grad_scale / accum) / self._scale_t
        finite = torch.isfinite(norm)
        if cfg.grad_clip > 0:
            mult = torch.clamp(cfg.grad_clip / (norm + 1e-6), max=1.0)
        else:
            mult = torch.ones_like(norm)
        step_mult = torch.where(finite, mult, torch.zeros_like(mult))
        out = step_mult / self._scale_t  # folds the unscale; 0.0 stays 0.0
        if self.dynamic_scale:
            good = torch.where(finite, self._good_t + 1.0,
                               torch.zeros_like(self._good_t))
            grow = good >= float(self._growth_interval)
            new_scale = torch.where(
                finite,
                torch.where(grow, (self._scale_t * 2.0).clamp(max=65536.0),
                            self._scale_t),
                (self._scale_t * 0.5).clamp(min=1.0))
            self._good_t.copy_(torch.where(grow, torch.zeros_like(good), good))
            self._scale_t.copy_(new_scale)
            self._skips_t += (~finite).float()
        return out

    def _grad_norm_and_clip_scale(self) -> float:
        """Global grad norm over the flat buckets; returns the extra scale
        factor that implements clipping (folded into the fused SGD's
        grad_scale — no separate pass over the grads).  Handles the fp16
        dynamic scaler (returns 0.0 => skip this step)."""
        cfg = self.cfg
        if not (cfg.grad_clip > 0 or self.dynamic_scale):
            return 1.0
        sq = None
        for b in self.reducer.buckets:
            n = b.flat.float().square().sum()
            sq = n if sq is None else sq + n
        # norm of the SUMMED (not yet averaged/unscaled) grads
        raw_norm = torch.sqrt(sq)
        accum = max(1, cfg.accum_steps)
        eff = self.reducer.grad_scale / (self.loss_scale * accum)  # avg+unscale
        norm = (raw_norm * eff).item()
        if self.dynamic_scale and not (norm == norm and norm < float("inf")):
            self.loss_scale = max(self.loss_scale / 2.0, 1.0)
            self._good_steps = 0
            return 0.0
        if self.dynamic_scale:
            self._good_steps += 1
            if self._good_steps >= self._growth_interval:
                self.loss_scale = min(self.loss_scale * 2.0, 65536.0)
                self._good_steps = 0
        if cfg.grad_clip > 0 and norm > cfg.grad_clip:
            return cfg.grad_clip / (norm + 1e-6)
        return 1.0

    # ------------------------------------------------------------------
    def _step_body(self, img, mask, depth, lr) -> torch.Tensor:
        """One full optimization step with NO host round-trips — the
        hipGraph capture region (also runs eagerly during graph warmup).
        ``lr`` may be a device scalar (graph replay updates it in place)."""
        self.reducer.zero_grad()
        logits = self.model(img, depth) if depth is not None else self.model(img)
        loss = self.criterion(logits, mask)
        if self._use_device_scaler and self.dynamic_scale:
            (loss * self._scale_t).backward()
        else:
            loss.backward()
        self.reducer.finalize()
        if self._use_device_scaler:
            mult = self._device_clip_and_scale()
            self.optimizer.step(lr, self.reducer.grad_scale, scale_t=mult)
        else:
            self.optimizer.step(lr, self.reducer.grad_scale)
        return loss

    def _step_graph(self, img, mask, depth) -> torch.Tensor:
        if self._graph_warm < 2:   # eager warmups stabilize the allocator
            self._graph_warm += 1
            loss = self._step_body(img, mask, depth,
                                   self._lr_at(self.iter_count))
            self.iter_count += 1
            return loss.detach()
        if self._graph is None:
            try:
                self._g_lr = torch.zeros((), device=self.device)
                self._g_img = img.clone()
                self._g_mask = mask.clone()
                self._g_depth = depth.clone() if depth is not None else None
                torch.cuda.synchronize()
                self._graph = torch.cuda.CUDAGraph()
                # thread_local: the loader's pin-memory thread calls
                # hipHostMalloc while we capture; in the default "global"
                # mode that INVALIDATES the capture (the real-loader soak
                # died with hipErrorStreamCaptureInvalidated).  The pin
                # thread does no device work, so relaxing is safe.
                with torch.cuda.graph(self._graph,
                                      capture_error_mode="thread_local"):
                    self._g_loss = self._step_body(self._g_img, self._g_mask,
                                                   self._g_depth, self._g_lr)
                torch.cuda.synchronize()
            except Exception as e:  # capture failure must not kill training
                self.log.info(f"hipGraph capture failed ({e!r}); "
                              f"falling back to eager steps")
                self._graph = None
                self._graph_ok = False
                self._h2d_pending = []
                loss = self._step_body(img, mask, depth,
                                       self._lr_at(self.iter_count))
                self.iter_count += 1
                return loss.detach()
        # Bound the host to ONE step ahead: without this, replay() lets the
        # host free-run and the loader RECYCLES its pinned staging buffers
        # before the async input copies execute — replays then train on
        # garbage (or DMA from freed pinned memory and fault).  The event
        # retires the PREVIOUS step's copies before this step's begin, and
        # _g_hold keeps the previous batch's buffers alive until then.
        if self._g_ev is None:
            self._g_ev = torch.cuda.Event()
        else:
            self._g_ev.synchronize()
        self._g_img.copy_(img, non_blocking=True)
        self._g_mask.copy_(mask, non_blocking=True)
        if depth is not None:
            self._g_depth.copy_(depth, non_blocking=True)
        self._g_ev.record()
        # hold BOTH the device tensors and the CPU(pinned) sources of this
        # step's H2D copies until the next fence retires them
        self._g_hold = (self._h2d_pending, img, mask, depth)
        self._h2d_pending = []
        self._g_lr.fill_(self._lr_at(self.iter_count))
        self._graph.replay()
        self.iter_count += 1
        return self._g_loss.detach()

    def step(self, img: torch.Tensor, mask: torch.Tensor,
             depth: Optional[torch.Tensor] = None) -> torch.Tensor:
        """One loader batch.  With ``accum_steps == 1`` (default) this is a
        full optimization step.  With accumulation, the first N-1 calls are
        no_sync micro-batches (grads accumulate into the bucket views, no
        collectives); the N-th runs the overlapped all-reduce of the
        accumulated sums + the fused SGD with 1/(world*accum) folded in.
        Returns the (detached, device-side) unscaled micro-batch loss."""
        if self._graph_ok:
            return self._step_graph(img, mask, depth)
        accum = max(1, self.cfg.accum_steps)
        first = self._accum_count == 0
        final = self._accum_count == accum - 1
        if first:
            self.reducer.zero_grad()
        logits = self.model(img, depth) if depth is not None else self.model(img)
        loss = self.criterion(logits, mask)
        if self._use_device_scaler and self.dynamic_scale:
            scaled = loss * self._scale_t  # device-resident scale, no sync
        elif self.loss_scale != 1.0:
            scaled = loss * self.loss_scale
        else:
            scaled = loss
        if final:
            scaled.backward()
        else:
            with self.reducer.no_sync():
                scaled.backward()
            self._accum_count += 1
            return loss.detach()
        self._accum_count = 0
        self.reducer.finalize()
        if self._use_device_scaler:
            mult_t = self._device_clip_and_scale()
            self.optimizer.step(self._lr_at(self.iter_count),
                                self.reducer.grad_scale / accum,
                                scale_t=mult_t)
        else:
            clip_scale = self._grad_norm_and_clip_scale()
            if clip_scale != 0.0:  # 0.0 == overflow: skip step, scale halved
                lr = self._lr_at(self.iter_count)
                self.optimizer.step(
                    lr, self.reducer.grad_scale / (self.loss_scale * accum)
                        * clip_scale)
            else:
                self.skipped_steps += 1
                self.log.info(f"fp16 overflow: step {self.iter_count} "
                              f"skipped, loss_scale -> {self.loss_scale} "
                              f"({self.skipped_steps} skipped total)")
        self.iter_count += 1
        return loss.detach()

    # ------------------------------------------------------------------
    def train(self) -> None:
        cfg = self.cfg
        loader, sampler = create_loader(cfg, "train", self.distributed,
                                        self.rank, self.world)
        accum = max(1, cfg.accum_steps)
        self.total_iters = max(1, cfg.epoch_num * len(loader) // accum)
        if cfg.resume and os.path.exists(self.paths["checkpoint_final"]):
            self.start_epoch, scaler = load_checkpoint(self.paths, self.model,
                                                       self.optimizer,
                                                       map_location=self.device)
            if scaler and self.dynamic_scale:
                self.loss_scale = float(scaler.get("loss_scale", self.loss_scale))
                if self._use_device_scaler:
                    self._scale_t.fill_(self.loss_scale)
            self.iter_count = self.start_epoch * len(loader) // accum
            self.log.info(f"resumed at epoch {self.start_epoch} "
                          f"(loss_scale {self.loss_scale})")
        if self.rank == 0 and cfg.tb:
            self.tb = TBRecorder(self.paths["tb"], enabled=True)

        from ..utils.profiler import profile_ctx
        self.model.train()
        loss_meter = AvgMeter()
        ips_meter = AvgMeter()
        prof_cm = profile_ctx(cfg.profile_steps,
                              os.path.join(self.paths["tb"], "profiler"),
                              self.rank)
        prof_step = prof_cm.__enter__()
        try:
            self._train_epochs(cfg, loader, sampler, loss_meter, ips_meter,
                               prof_step)
        finally:
            prof_cm.__exit__(None, None, None)
            if self.tb is not None:
                self.tb.close()

    def _train_epochs(self, cfg, loader, sampler, loss_meter, ips_meter,
                      prof_step) -> None:
        for epoch in range(self.start_epoch, cfg.epoch_num):
            if sampler is not None:
                sampler.set_epoch(epoch)
            loader.dataset.set_epoch(epoch)
            t_last = time.perf_counter()
            win_imgs = 0
            for i, batch in enumerate(loader):
                if len(batch) == 4:
                    img, depth, mask, _ = batch
                    depth = self._to_device(depth, cast=True)
                else:
                    img, mask, _ = batch
                    depth = None
                img = self._to_device(img, cast=True)
                mask = self._to_device(mask, cast=False)
                loss = self.step(img, mask, depth)
                prof_step()
                win_imgs += img.shape[0] * self.world
                if (i + 1) % cfg.log_interval == 0 or i + 1 == len(loader):
                    # .item() syncs to the last step's completion, so the
                    # window wall-clock is honest even under hipGraph replay
                    # (per-step timing would measure async enqueue rates)
                    lv = loss.item()
                    now = time.perf_counter()
                    ips_meter.update(win_imgs / (now - t_last))
                    t_last = now
                    win_imgs = 0
                    loss_meter.update(lv)
                    comm = ""
                    if self.reducer.stats_mode and self.distributed:
                        st = self.reducer.pop_stats()
                        comm = (f" comm[buckets {st['launches']} spread "
                                f"{st['launch_spread_ms']:.1f}ms wait "
                                f"{st['wait_ms']:.1f}ms]")
                    scal = ""
                    if self.dynamic_scale:
                        # host sync amortized to the log interval only
                        ls, sk = self._scaler_state()
                        scal = f" scale {ls:.0f}"
                        if sk:
                            scal += f" (skipped {sk})"
                    lr_it = max(0, self.iter_count - 1)
                    self.log.info(
                        f"ep {epoch} it {i + 1}/{len(loader)} "
                        f"loss {loss_meter.val:.4f} (avg {loss_meter.avg:.4f}) "
                        f"lr {self._lr_at(lr_it):.5f} "
                        f"imgs/s {ips_meter.avg:.1f}{scal}{comm}")
                    if self.tb is not None:
                        self.tb.scalar("train/loss", loss_meter.val, self.iter_count)
                        self.tb.scalar("train/lr", self._lr_at(lr_it),
                                       self.iter_count)
            if self.tb is not None:   # reference TB parity: image panel per epoch
                self.model.eval()     # don't touch BN running stats (no SyncBN
                with torch.no_grad():  # collective in eval: rank-0-only is safe)
                    pred = torch.sigmoid(
                        (self.model(img, depth) if depth is not None
                         else self.model(img))[:1].float())
                self.model.train()
                self.tb.image("train/img", img[0].float().cpu(), epoch)
                self.tb.image("train/mask", mask[0].float().cpu(), epoch)
                self.tb.image("train/pred", pred[0].cpu(), epoch)
            if self.rank == 0:
                save_checkpoint(self.paths, epoch, self.model, self.optimizer,
                                {"loss_scale": self._scaler_state()[0]})
            self._barrier()

    # ------------------------------------------------------------------
    @torch.no_grad()
    def test(self, save_png: bool = True, max_png: int = 16) -> dict:
        """Distributed eval on the fixed-seed synthetic val split (SURVEY
        §3.2): MAE + F-measure with cross-rank reduction, PNG saves."""
        cfg = self.cfg
        loader, _ = create_loader(cfg, "val", self.distributed,
                                  self.rank, self.world)
        self.model.eval()
        mae = MAEMeter()
        prf = PRFMeter()
        n_png = 0
        pre_dir = os.path.join(self.paths["pre"], "synthetic_val")
        if self.rank == 0 and save_png:
            os.makedirs(pre_dir, exist_ok=True)
        for batch in loader:
            if len(batch) == 4:
                img, depth, mask, names = batch
                depth = self._to_device(depth, cast=True)
            else:
                img, mask, names = batch
                depth = None
            img = self._to_device(img, cast=True)
            mask = self._to_device(mask, cast=False)
            logits = self.model(img, depth) if depth is not None else self.model(img)
            pred = torch.sigmoid(logits.float())
            mae.update(pred, mask)
            prf.update(pred, mask)
            if self.rank == 0 and save_png and n_png < max_png:
                for b in range(min(pred.shape[0], max_png - n_png)):
                    write_gray_png(os.path.join(pre_dir, f"{names[b]}.png"),
                                   pred[b, 0].cpu().numpy())
                    n_png += 1
        res = {"mae": mae.compute(self.distributed),
               **{k: v for k, v in prf.compute(self.distributed).items()
                  if k in ("maxf", "meanf")}}
        if self.rank == 0:
            rec = ResultsRecorder(self.paths["root"])
            rec.add("synthetic_val", res)
            rec.save()
            self.log.info(f"eval: MAE {res['mae']:.4f} maxF {res['maxf']:.4f} "
                          f"meanF {res['meanf']:.4f}")
        self._barrier()
        self.model.train()
        return res

    # ------------------------------------------------------------------
    def _scaler_state(self) -> tuple[float, int]:
        """(loss_scale, skipped_steps) — reads the device scaler state (one
        host sync; call only at log/checkpoint points, never per step)."""
        if self._use_device_scaler and self.dynamic_scale:
            self.loss_scale = float(self._scale_t.item())
            self.skipped_steps = int(self._skips_t.item())
        return self.loss_scale, self.skipped_steps

    # ------------------------------------------------------------------
    def _barrier(self) -> None:
        if self.distributed and dist.is_initialized():
            dist.barrier()
