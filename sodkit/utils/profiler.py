"""torch.profiler wrapper: `--profile_steps N` wraps N train
steps and writes a chrome trace under the experiment tree.  rocprofv3 remains
the tool for per-kernel HW counters (tools/prof_summary.py,
tools/pmc_summary.py)."""

from __future__ import annotations

import contextlib
import os


@contextlib.contextmanager
def profile_ctx(n_steps: int, out_dir: str, rank: int = 0):
    """Yields a callable to invoke once per step; profiles steps [1, n]."""
    if n_steps <= 0 or rank != 0:
        yield lambda: None
        return
    import torch.profiler as tp
    os.makedirs(out_dir, exist_ok=True)
    prof = tp.profile(
        activities=[tp.ProfilerActivity.CPU, tp.ProfilerActivity.CUDA],
        schedule=tp.schedule(wait=1, warmup=1, active=max(1, n_steps - 2)),
        on_trace_ready=tp.tensorboard_trace_handler(out_dir),
    )
    prof.start()
    try:
        yield prof.step
    finally:
        prof.stop()
