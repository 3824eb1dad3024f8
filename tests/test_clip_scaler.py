"""Grad clipping (folded into the SGD grad scale) and the fp16 dynamic loss
scaler — CPU tests via the Solver step path."""

import torch

from sodkit.config import make_config
from sodkit.engine import Solver


def _solver(**kw):
    cfg = make_config("cpu_plumbing", size=32, train_samples=2, val_samples=2,
                      num_workers=0, **kw)
    return Solver(cfg)


def test_grad_clip_limits_update():
    torch.manual_seed(0)
    g = torch.Generator().manual_seed(1)
    x = torch.rand(1, 3, 32, 32, generator=g)
    t = (torch.rand(1, 1, 32, 32, generator=g) > 0.5).float()

    s_clip = _solver(grad_clip=1e-6, lr=1.0, lr_type="const", weight_decay=0.0,
                     output_root="/tmp/sodclip1")
    before = [p.detach().clone() for p in s_clip.model.parameters()]
    s_clip.step(x, t)
    moved = sum((p.detach() - b).abs().max().item()
                for p, b in zip(s_clip.model.parameters(), before))
    # with clip 1e-6 and lr 1 the update is ~clip-sized (plus weight decay)
    assert moved < 1e-2, moved

    s_noclip = _solver(grad_clip=0.0, lr=1.0, lr_type="const", weight_decay=0.0,
                       output_root="/tmp/sodclip2")
    before = [p.detach().clone() for p in s_noclip.model.parameters()]
    s_noclip.step(x, t)
    moved_nc = sum((p.detach() - b).abs().max().item()
                   for p, b in zip(s_noclip.model.parameters(), before))
    assert moved_nc > moved * 10


def test_dynamic_scaler_skips_on_overflow():
    s = _solver(output_root="/tmp/sodscal")
    s.dynamic_scale = True
    s.loss_scale = 64.0
    # poison a bucket with inf to simulate overflow
    g = torch.Generator().manual_seed(2)
    x = torch.rand(1, 3, 32, 32, generator=g)
    t = (torch.rand(1, 1, 32, 32, generator=g) > 0.5).float()
    before = [p.detach().clone() for p in s.model.parameters()]

    orig_finalize = s.reducer.finalize

    def poisoned():
        orig_finalize()
        s.reducer.buckets[0].flat[0] = float("inf")

    s.reducer.finalize = poisoned
    s.step(x, t)
    # step skipped, scale halved
    assert s.loss_scale == 32.0
    for p, b in zip(s.model.parameters(), before):
        assert torch.equal(p.detach(), b)
    s.reducer.finalize = orig_finalize
    s.step(x, t)  # recovers
    assert s._good_steps == 1


def test_device_scaler_state_machine():
    """The GPU sync-free scaler's tensor state machine, exercised on CPU
    tensors (identical torch ops; only the fused-SGD consumption is GPU)."""
    s = _solver(output_root="/tmp/sodscal3", grad_clip=0.0)
    s.dynamic_scale = True
    s._use_device_scaler = True
    s._growth_interval = 2
    s._scale_t = torch.tensor(64.0)
    s._good_t = torch.zeros(())
    s._skips_t = torch.zeros(())
    g = torch.Generator().manual_seed(2)
    x = torch.rand(1, 3, 32, 32, generator=g)
    t = (torch.rand(1, 1, 32, 32, generator=g) > 0.5).float()
    s.reducer.zero_grad()
    (s.criterion(s.model(x), t) * s._scale_t).backward()
    s.reducer.finalize()

    mult = s._device_clip_and_scale()
    assert abs(mult.item() - 1 / 64.0) < 1e-9   # unscale folded in
    assert s._scale_t.item() == 64.0 and s._good_t.item() == 1

    # second good step hits the growth interval: scale doubles, counter resets
    s._device_clip_and_scale()
    assert s._scale_t.item() == 128.0 and s._good_t.item() == 0

    # overflow: multiplier 0.0 (skip), scale halves, skip counted
    s.reducer.buckets[0].flat[0] = float("inf")
    mult = s._device_clip_and_scale()
    assert mult.item() == 0.0
    assert s._scale_t.item() == 64.0
    assert s._skips_t.item() == 1 and s._good_t.item() == 0
    ls, sk = s._scaler_state()
    assert ls == 64.0 and sk == 1

    # grad_clip composes: tiny clip forces mult < 1/scale
    s.reducer.buckets[0].flat[0] = 0.0
    s.cfg.grad_clip = 1e-8
    mult = s._device_clip_and_scale()
    assert 0.0 < mult.item() < 1 / 64.0
