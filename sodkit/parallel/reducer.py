"""GradReducer — the hand-built data-parallel gradient engine (replaces
torch.nn.parallel.DistributedDataParallel).

Design, MI355X-first:
  * Parameters are grouped in REVERSE registration order (≈ backward arrival
    order) into flat buckets (default 25 MB, sized for RCCL over xGMI — with
    ~2 ms of total gradient traffic per step on a fully connected 8-GPU node,
    overlap correctness matters more than bucket heroics).
  * Each param's ``.grad`` IS a view into its bucket slot — autograd
    accumulates straight into the flat buffer; no copy, no regather
    (gradient_as_bucket_view by construction).
  * ``register_post_accumulate_grad_hook`` counts arrivals; when a bucket is
    complete its all-reduce launches immediately on a SIDE HIP stream (the
    comm stream waits on an event recorded on the compute stream), so
    communication overlaps the rest of backward.
  * ``finalize()`` makes the compute stream wait for every in-flight
    collective.  The 1/world_size average is NOT applied here — it folds
    into the fused SGD's grad_scale (one fewer pass over the grads).
  * On CPU (gloo) the same engine runs with the stream layer inert — one
    code path, the [BJ:7]/multi-process-CPU tests exercise exactly the logic
    the GPU runs.

Init-time C2 broadcast of rank-0 parameters is provided (`broadcast_params`).
An ASSERT mode (SODKIT_COMM_ASSERT=1) checksums buckets around each
all-reduce (SURVEY §5.2 race detection).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


def _world(pg) -> int:
    if dist.is_initialized():
        return dist.get_world_size(pg)
    return 1


class _Bucket:
    __slots__ = ("params", "flat", "arrived", "work", "ready_event", "index",
                 "launched")

    def __init__(self, index: int):
        self.index = index
        self.params: list[torch.nn.Parameter] = []
        self.flat: Optional[torch.Tensor] = None
        self.arrived = 0
        self.work = None
        self.ready_event = None
        self.launched = False  # all-reduce issued this step (survives the
        #                        assert-mode synchronous wait, which nulls
        #                        `work` — finalize() must not re-launch)


class GradReducer:
    def __init__(self, model: torch.nn.Module, process_group=None,
                 bucket_mb: int = 25, overlap: bool = True):
        self.pg = process_group
        self.overlap = overlap and torch.cuda.is_available()
        self._sync = True            # False inside no_sync() micro-batches
        self.assert_mode = os.environ.get("SODKIT_COMM_ASSERT", "0") == "1"
        params = [p for p in model.parameters() if p.requires_grad]
        params.reverse()  # reverse registration order ≈ backward arrival order
        self._params = params

        # --- bucket assignment (per dtype: bf16 conv weights and fp32 BN
        # params flatten into separate buckets; arrival order within each
        # dtype group still tracks backward order) ---
        bucket_bytes = bucket_mb * (1 << 20)
        self.buckets: list[_Bucket] = []
        by_dtype: dict[torch.dtype, list[torch.nn.Parameter]] = {}
        for p in params:
            by_dtype.setdefault(p.dtype, []).append(p)
        for plist in by_dtype.values():
            cur = _Bucket(len(self.buckets))
            cur_bytes = 0
            for p in plist:
                nbytes = p.numel() * p.element_size()
                if cur.params and cur_bytes + nbytes > bucket_bytes:
                    self.buckets.append(cur)
                    cur = _Bucket(len(self.buckets))
                    cur_bytes = 0
                cur.params.append(p)
                cur_bytes += nbytes
            if cur.params:
                self.buckets.append(cur)

        # --- flat storage + grad views ---
        self._p2bucket: dict[torch.nn.Parameter, _Bucket] = {}
        self.grad_views: dict[torch.nn.Parameter, torch.Tensor] = {}
        for b in self.buckets:
            total = sum(p.numel() for p in b.params)
            dev = b.params[0].device
            dt = b.params[0].dtype
            b.flat = torch.zeros(total, device=dev, dtype=dt)
            off = 0
            for p in b.params:
                p.grad = b.flat[off:off + p.numel()].view_as(p)
                self.grad_views[p] = p.grad
                off += p.numel()
                self._p2bucket[p] = b

        # --- deferred-grad mode (single-GPU): AccumulateGrad's per-param
        # add_ into the bucket views is ~280 launches of <=512 threads per
        # step at the ~4.5us launch floor (profiles/r2/
        # kernel_summary_close.txt, ~3.5% of the step).  At world==1 there
        # is no overlap to preserve, so instead: zero_grad() sets .grad=None
        # (autograd then ASSIGNS fresh grad tensors — zero kernels) and
        # finalize() lands them in the flats with ONE multicopy kernel.
        # Bit-identical to the view path (assignment == add-into-zero; the
        # gather is an exact copy).  Accumulation-safe: micro-steps 2..k add
        # into the live .grad and the gather copies the accumulated value.
        # The fused SGD keeps reading the stable bucket views (grad_views).
        # SODKIT_DEFER_GRADS=0 disables (A/B).
        self.defer = (_world(self.pg) == 1 and len(params) > 0
                      and params[0].is_cuda
                      and os.environ.get("SODKIT_DEFER_GRADS", "1") == "1")
        self._mc = None          # lazily built static copy tables
        self._ptr_cache = None   # last step's grad pointers (skip H2D)
        self._cache_captured = False

        self.comm_stream = (torch.cuda.Stream()
                            if self.overlap and _world(self.pg) > 1 else None)
        # per-bucket comm-overlap stats: SODKIT_COMM_STATS=1
        self.stats_mode = os.environ.get("SODKIT_COMM_STATS", "0") == "1"
        self.stats = {"launches": 0, "launch_t": [], "wait_ms": 0.0}
        self._hooks = []
        for p in params:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._on_grad_ready))

    # ------------------------------------------------------------------
    def broadcast_params(self, model: torch.nn.Module) -> None:
        """C2: rank-0 params + buffers to all (belt-and-braces on top of the
        identical-seed init)."""
        if _world(self.pg) <= 1:
            return
        for t in list(model.parameters()) + list(model.buffers()):
            dist.broadcast(t.data, src=0, group=self.pg)

    # ------------------------------------------------------------------
    def no_sync(self):
        """Context manager: backwards inside skip the bucket all-reduce
        (gradient-accumulation micro-batches; grads keep accumulating into
        the bucket views).  Same semantics as DDP.no_sync()."""
        import contextlib

        @contextlib.contextmanager
        def cm():
            prev = self._sync
            self._sync = False
            try:
                yield
            finally:
                self._sync = prev
        return cm()

    # ------------------------------------------------------------------
    def _on_grad_ready(self, p: torch.nn.Parameter) -> None:
        b = self._p2bucket[p]
        b.arrived += 1
        if b.arrived == len(b.params):
            b.arrived = 0          # cycle per backward (accumulation-safe)
            if self._sync:
                self._launch(b)

    def _launch(self, b: _Bucket) -> None:
        if _world(self.pg) <= 1:
            return
        b.launched = True
        if self.stats_mode:
            import time
            self.stats["launches"] += 1
            self.stats["launch_t"].append(time.perf_counter())
        if self.assert_mode:
            before = b.flat.float().sum().item()
        if self.comm_stream is not None:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())
            with torch.cuda.stream(self.comm_stream):
                self.comm_stream.wait_event(ev)
                b.work = dist.all_reduce(b.flat, group=self.pg, async_op=True)
        else:
            b.work = dist.all_reduce(b.flat, group=self.pg, async_op=True)
        if self.assert_mode:
            if b.work is not None:
                b.work.wait()
                b.work = None
            after = b.flat.float().sum().item()
            w = _world(self.pg)
            if not (abs(after) <= abs(before) * w * 4 + 1e3):  # sanity bound
                raise RuntimeError(
                    f"comm assert: bucket {b.index} sum {before} -> {after} "
                    f"not plausible for a {w}-way sum")

    # ------------------------------------------------------------------
    def _defer_gather(self) -> None:
        """One-kernel gather of every autograd-assigned .grad into its
        bucket slot (deferred-grad mode).  Segment sizes and destinations
        are static (bucket layout); only the source pointers change —
        and with the caching allocator they are step-stable, so the tiny
        H2D pointer refresh is skipped in steady state.  hipGraph-safe:
        under capture the refresh goes through a pinned staging buffer
        (async H2D only), and replays re-run the baked copy+kernel against
        the capture pool's stable addresses."""
        from ..ops._backend import ext

        if self._mc is None:
            chunk = int(ext().multicopy_chunk_bytes())
            plist: list[torch.nn.Parameter] = []
            dst_ptrs: list[int] = []
            sizes: list[int] = []
            for b in self.buckets:
                esz = b.params[0].element_size()
                base = b.flat.data_ptr()
                off = 0
                for p in b.params:
                    plist.append(p)
                    dst_ptrs.append(base + off * esz)
                    sizes.append(p.numel() * esz)
                    off += p.numel()
            dev = self.buckets[0].flat.device
            segmap, chunkmap = [], []
            for i, nb in enumerate(sizes):
                for c in range((nb + chunk - 1) // chunk):
                    segmap.append(i)
                    chunkmap.append(c)
            self._mc = (
                plist, dst_ptrs,
                torch.tensor(dst_ptrs, dtype=torch.int64, device=dev),
                torch.tensor(sizes, dtype=torch.int64, device=dev),
                torch.tensor(segmap, dtype=torch.int32, device=dev),
                torch.tensor(chunkmap, dtype=torch.int32, device=dev),
                torch.empty(len(plist), dtype=torch.int64, device=dev),
                torch.empty(len(plist), dtype=torch.int64, pin_memory=True),
            )
        (plist, dst_ptrs, dst_dev, nb_dev, segmap, chunkmap, src_dev,
         src_pin) = self._mc
        ptrs = []
        for i, p in enumerate(plist):
            g = p.grad
            if g is not None and g.is_contiguous() and g.dtype == p.dtype:
                ptrs.append(g.data_ptr())
            else:
                v = self.grad_views[p]
                if g is None:
                    v.detach().zero_()     # param got no grad this step
                else:                      # odd layout: land it directly
                    v.detach().copy_(g.reshape(v.shape).to(p.dtype))
                ptrs.append(dst_ptrs[i])   # self-copy no-op in the kernel
        capturing = torch.cuda.is_current_stream_capturing()
        if ptrs != self._ptr_cache or (self._cache_captured and not capturing):
            # second clause: a FAILED capture may have cached pointers whose
            # device-side refresh never executed — eager must re-copy
            self._ptr_cache = ptrs
            self._cache_captured = capturing
            t = torch.tensor(ptrs, dtype=torch.int64)
            if capturing:
                src_pin.copy_(t)
                src_dev.copy_(src_pin, non_blocking=True)
            else:
                src_dev.copy_(t)  # blocking: stream-ordered after last gather
        ext().multicopy(src_dev, dst_dev, nb_dev, segmap, chunkmap)
        # restore the public contract: after the gather, .grad IS the bucket
        # view again (host-only pointer stores).  Any consumer that bakes
        # .grad pointers — the fused SGD's tables, user code — sees the
        # STABLE views, not step-transient autograd tensors.  (Dropping the
        # last ref to an autograd grad here is safe: the caching allocator
        # is stream-ordered, so any reuse is enqueued after the gather.)
        for p in plist:
            p.grad = self.grad_views[p]

    # ------------------------------------------------------------------
    def finalize(self) -> None:
        """Call after the last (sync) backward of the step: wait for every
        in-flight collective (and launch any straggler buckets — a bucket
        whose params got no grad this backward has arrived != 0 ...== 0 via
        the cycling counter only when it completed)."""
        if self.defer:
            self._defer_gather()
            for b in self.buckets:
                b.arrived = 0
                b.launched = False
            return
        for b in self.buckets:
            if not b.launched and _world(self.pg) > 1 and self._sync:
                # straggler: either some param got no grad this step
                # (arrived != 0) or the bucket completed during a no_sync
                # micro-batch — its slot holds the accumulated sum,
                # all-reduce now so ranks agree.  (`launched`, not
                # `work is None`: assert mode waits synchronously and nulls
                # `work`, which must not look like a straggler.)
                self._launch(b)
            if b.work is not None:
                if self.stats_mode:
                    import time
                    t0 = time.perf_counter()
                    b.work.wait()
                    self.stats["wait_ms"] += (time.perf_counter() - t0) * 1e3
                else:
                    b.work.wait()  # current stream waits for the collective
                b.work = None
            b.arrived = 0
            b.launched = False

    def zero_grad(self) -> None:
        if self.defer:
            # no memsets: .grad=None makes autograd ASSIGN the next grads
            # (zero kernels) and the finalize gather overwrites every slot
            for p in self._params:
                p.grad = None
            for b in self.buckets:
                b.arrived = 0
                b.launched = False
            return
        for b in self.buckets:
            b.flat.zero_()
            b.arrived = 0
            b.launched = False

    def pop_stats(self) -> dict:
        """Comm-overlap summary since the last call (stats mode only):
        bucket launch spread within backward + host-side wait time."""
        st = self.stats
        out = {"launches": st["launches"],
               "launch_spread_ms": ((max(st["launch_t"]) - min(st["launch_t"]))
                                    * 1e3 if len(st["launch_t"]) > 1 else 0.0),
               "wait_ms": st["wait_ms"]}
        self.stats = {"launches": 0, "launch_t": [], "wait_ms": 0.0}
        return out

    @property
    def grad_scale(self) -> float:
        """Fold this into the optimizer's grad_scale (× any loss-scale)."""
        return 1.0 / _world(self.pg)

    def detach(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
