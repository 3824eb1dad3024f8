"""Utility-layer unit tests: PNG writer, recorders, meters."""
import os

import numpy as np
import torch

from sodkit.utils import AvgMeter, Timer
from sodkit.utils.png import write_gray_png
from sodkit.utils.recorder import ResultsRecorder, TBRecorder


def test_png_writer_roundtrippable(tmp_path):
    arr = np.linspace(0, 1, 64 * 48).reshape(64, 48).astype(np.float32)
    path = str(tmp_path / "t.png")
    write_gray_png(path, arr)
    data = open(path, "rb").read()
    assert data[:8] == b"\x89PNG\r\n\x1a\n"
    assert b"IHDR" in data and b"IDAT" in data and b"IEND" in data


def test_results_recorder_csv(tmp_path):
    rec = ResultsRecorder(str(tmp_path))
    rec.add("dsA", {"mae": 0.1, "maxf": 0.9})
    rec.add("dsB", {"mae": 0.2, "maxf": 0.8})
    rec.save()
    lines = open(tmp_path / "results.csv").read().strip().split("\n")
    assert lines[0] == "dataset,mae,maxf" and len(lines) == 3


def test_tb_recorder_degrades_gracefully(tmp_path):
    tb = TBRecorder(str(tmp_path), enabled=True)  # tensorboard absent -> noop
    tb.scalar("x", 1.0, 0)
    tb.close()


def test_meters():
    m = AvgMeter()
    for v in (1.0, 2.0, 3.0):
        m.update(v)
    assert m.avg == 2.0 and m.val == 3.0
    t = Timer()
    assert t.elapsed() >= 0.0 and t.lap() >= 0.0
