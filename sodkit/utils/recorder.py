"""Observability recorders — parity with the reference's utils/recorder.py
(TBRecorder, Timer, XLSXRecorder).  TensorBoard is an
optional dependency (absent in this image: the recorder degrades to a no-op
with a one-line notice); results tables always write CSV and additionally
XLSX when an engine is importable."""

from __future__ import annotations

import csv
import os


class TBRecorder:
    def __init__(self, logdir: str, enabled: bool = True):
        self.writer = None
        os.makedirs(logdir, exist_ok=True)
        if not enabled:
            return
        try:
            from torch.utils.tensorboard import SummaryWriter  # type: ignore
            self.writer = SummaryWriter(logdir)
        except Exception:
            print("[sodkit] tensorboard unavailable; TB recording disabled")

    def scalar(self, tag: str, value: float, step: int) -> None:
        if self.writer is not None:
            self.writer.add_scalar(tag, value, step)

    def image(self, tag: str, img, step: int) -> None:
        """img: [C,H,W] or [H,W] tensor in [0,1] (reference TB parity:
        periodic input/pred/mask image panels)."""
        if self.writer is not None:
            if img.ndim == 2:
                img = img.unsqueeze(0)
            self.writer.add_image(tag, img.float().clamp(0, 1), step)

    def close(self) -> None:
        if self.writer is not None:
            self.writer.close()


class ResultsRecorder:
    """Per-dataset metric table -> results.csv (+ results.xlsx if possible)."""

    def __init__(self, out_dir: str):
        self.out_dir = out_dir
        self.rows: list[dict] = []

    def add(self, dataset: str, metrics: dict) -> None:
        self.rows.append({"dataset": dataset, **metrics})

    def save(self) -> None:
        if not self.rows:
            return
        os.makedirs(self.out_dir, exist_ok=True)
        keys = list(self.rows[0].keys())
        with open(os.path.join(self.out_dir, "results.csv"), "w", newline="") as f:
            w = csv.DictWriter(f, fieldnames=keys)
            w.writeheader()
            for r in self.rows:
                w.writerow(r)
        try:
            import pandas as pd  # xlsx parity when an engine exists
            pd.DataFrame(self.rows).to_excel(
                os.path.join(self.out_dir, "results.xlsx"), index=False)
        except Exception:
            pass
