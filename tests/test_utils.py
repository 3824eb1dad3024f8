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
    # results.xlsx is written unconditionally by the dependency-free writer
    import zipfile
    xp = os.path.join(str(tmp_path), "results.xlsx")
    assert os.path.exists(xp)
    with zipfile.ZipFile(xp) as z:
        assert z.testzip() is None
        sheet = z.read("xl/worksheets/sheet1.xml").decode()
    assert "dsA" in sheet and "0.9" in sheet


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


def test_timer_laps():
    import time
    from sodkit.utils import Timer
    t = Timer()
    time.sleep(0.01)
    l1 = t.lap()
    assert 0.005 < l1 < 1.0
    l2 = t.lap()
    assert l2 < l1            # immediate second lap is shorter
    assert t.elapsed() >= l1


def test_loader_with_workers():
    """num_workers > 0: the seeded generator must stay deterministic across
    worker processes (each index derives its own generator)."""
    from sodkit.config import make_config
    from sodkit.data import create_loader
    cfg = make_config("cpu_plumbing", size=32, train_samples=8, batch_size=4,
                      num_workers=2)
    l1, _ = create_loader(cfg, "val", distributed=False)
    l2, _ = create_loader(cfg, "val", distributed=False)
    import torch
    b1 = next(iter(l1))
    b2 = next(iter(l2))
    assert torch.equal(b1[0], b2[0]) and torch.equal(b1[1], b2[1])


def test_tb_recorder_image_noop_without_tb(tmp_path):
    import torch
    from sodkit.utils.recorder import TBRecorder
    r = TBRecorder(str(tmp_path), enabled=True)  # tb not installed -> no-op
    r.image("x", torch.rand(3, 8, 8), 0)         # must not raise
    r.image("y", torch.rand(8, 8), 1)            # 2D promoted to [1,H,W]
    r.scalar("z", 1.0, 0)
    r.close()
