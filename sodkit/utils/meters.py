"""AvgMeter / Timer — parity with the reference's utils/misc.py + recorder Timer."""

from __future__ import annotations

import time


class AvgMeter:
    """Running average of a scalar (loss smoothing, imgs/sec, ...)."""

    def __init__(self) -> None:
        self.reset()

    def reset(self) -> None:
        self.val = 0.0
        self.sum = 0.0
        self.count = 0
        self.avg = 0.0

    def update(self, val: float, n: int = 1) -> None:
        self.val = float(val)
        self.sum += float(val) * n
        self.count += n
        self.avg = self.sum / max(self.count, 1)

    def __repr__(self) -> str:  # pragma: no cover
        return f"AvgMeter(val={self.val:.4g}, avg={self.avg:.4g}, n={self.count})"


class Timer:
    """Wall-clock stopwatch with lap support."""

    def __init__(self) -> None:
        self.start()

    def start(self) -> None:
        self.t0 = time.perf_counter()
        self.last = self.t0

    def lap(self) -> float:
        now = time.perf_counter()
        dt = now - self.last
        self.last = now
        return dt

    def elapsed(self) -> float:
        return time.perf_counter() - self.t0
