"""The [BJ:7] CPU plumbing config end-to-end: config -> trainer -> model ->
loss -> ckpt -> resume -> test."""

import os

import pytest
import torch

from sodkit.config import make_config
from sodkit.engine import Solver
from sodkit.engine.checkpoint import exp_paths, load_checkpoint


@pytest.fixture()
def cfg(tmp_path):
    return make_config("cpu_plumbing", output_root=str(tmp_path), size=48,
                       train_samples=4, val_samples=2, epoch_num=1,
                       num_workers=0, log_interval=1)


@pytest.mark.timeout(600)
def test_train_then_test_writes_reference_layout(cfg):
    s = Solver(cfg)
    s.train()
    res = s.test()
    p = exp_paths(cfg.output_root, cfg.exp_name)
    assert os.path.exists(p["state_final"])
    assert os.path.exists(p["checkpoint_final"])
    assert os.path.exists(p["cfg"])
    assert os.path.exists(os.path.join(p["root"], "results.csv"))
    pre = os.path.join(p["pre"], "synthetic_val")
    assert os.path.isdir(pre) and len(os.listdir(pre)) > 0
    assert 0 <= res["mae"] <= 1
    # the deploy artifact round-trips strict into a fresh model
    from sodkit.models import build_model
    state = torch.load(p["state_final"], map_location="cpu",
                       weights_only=True)
    fresh = build_model(cfg.model)
    fresh.load_state_dict(state, strict=True)
    for (n1, p1), (n2, p2) in zip(s.model.state_dict().items(),
                                  fresh.state_dict().items()):
        assert n1 == n2 and torch.equal(p1, p2)


@pytest.mark.timeout(600)
def test_loss_decreases_on_fixed_batch(cfg):
    s = Solver(cfg.replace(lr=0.02, lr_type="const"))
    g = torch.Generator().manual_seed(5)
    x = torch.rand(2, 3, 48, 48, generator=g)
    t = (torch.rand(2, 1, 48, 48, generator=g) > 0.5).float()
    s.total_iters = 1000
    first = s.step(x, t).item()
    for _ in range(24):
        last = s.step(x, t).item()
    assert last < first * 0.8, (first, last)


@pytest.mark.timeout(600)
def test_checkpoint_resume_bitwise(cfg):
    s = Solver(cfg)
    s.train()  # writes epoch-0 checkpoint
    p = exp_paths(cfg.output_root, cfg.exp_name)

    s2 = Solver(cfg.replace(resume=True))
    next_epoch, scaler_state = load_checkpoint(p, s2.model, s2.optimizer)
    assert next_epoch == 1
    assert scaler_state is not None and "loss_scale" in scaler_state
    # params and optimizer masters bitwise equal after resume
    for a, b in zip(s.model.parameters(), s2.model.parameters()):
        assert torch.equal(a, b)
    for pa, pb in zip(s.optimizer.params(), s2.optimizer.params()):
        assert torch.equal(s.optimizer.state[pa]["momentum_buffer"],
                           s2.optimizer.state[pb]["momentum_buffer"])
        assert torch.equal(s.optimizer.state[pa]["master"],
                           s2.optimizer.state[pb]["master"])


@pytest.mark.timeout(600)
def test_poly_lr_schedule_with_warmup(cfg):
    s = Solver(cfg.replace(lr=0.1, lr_type="poly", lr_power=0.9,
                           warmup_iters=4))
    s.total_iters = 20
    # linear warmup: (it+1)/warmup_iters
    assert s._lr_at(0) == pytest.approx(0.1 * 1 / 4)
    assert s._lr_at(3) == pytest.approx(0.1 * 4 / 4)
    # poly decay after warmup
    for it in (4, 10, 19):
        assert s._lr_at(it) == pytest.approx(0.1 * (1 - it / 20) ** 0.9)
    # last iter keeps a positive LR (frac capped at (total-1)/total)
    assert s._lr_at(25) == pytest.approx(0.1 * (1 - 19 / 20) ** 0.9)
    # const fallback
    s2 = Solver(cfg.replace(lr=0.05, lr_type="const", warmup_iters=0))
    assert s2._lr_at(0) == s2._lr_at(999) == 0.05


@pytest.mark.timeout(600)
def test_resume_restores_dynamic_loss_scale(cfg, tmp_path):
    # fp16 dynamic-scaler state must survive a resume (CPU: dtype stays
    # fp32 compute but the scaler bookkeeping is dtype-driven)
    c = cfg.replace(dtype="fp16", loss_scale=256.0, exp_name="scaler_resume")
    s = Solver(c)
    s.train()
    s.loss_scale = 64.0  # pretend overflows halved it mid-run
    from sodkit.engine.checkpoint import save_checkpoint
    p = exp_paths(c.output_root, c.exp_name)
    save_checkpoint(p, 0, s.model, s.optimizer, {"loss_scale": s.loss_scale})

    s2 = Solver(c.replace(resume=True, epoch_num=2))
    s2.train()  # resumes: must pick up loss_scale=64, then run epoch 1
    assert s2.loss_scale <= 64.0  # restored (may halve further on overflow)
    assert s2.start_epoch == 1


@pytest.mark.timeout(600)
def test_profiler_writes_trace(cfg):
    s = Solver(cfg.replace(profile_steps=3, exp_name="prof"))
    s.train()
    tb_dir = exp_paths(cfg.output_root, "prof")["tb"]
    prof_dir = os.path.join(tb_dir, "profiler")
    assert os.path.isdir(prof_dir)
    files = os.listdir(prof_dir)
    assert any("pt.trace" in f or f.endswith(".json") for f in files), files


def test_reducer_detach_releases_hooks():
    import torch as t
    from sodkit.parallel import GradReducer
    m = t.nn.Linear(4, 4)
    red = GradReducer(m, bucket_mb=1)
    x = t.ones(2, 4)
    m(x).sum().backward()
    red.finalize()
    g1 = m.weight.grad.clone()
    red.detach()
    red.zero_grad()
    m(x).sum().backward()     # hooks gone: plain accumulation into the views
    assert t.allclose(m.weight.grad, g1)  # same grads, no reducer involvement
    assert not red._hooks


@pytest.mark.timeout(600)
def test_solver_grad_accumulation(cfg):
    c = cfg.replace(accum_steps=2, train_samples=8, exp_name="accum")
    s = Solver(c)
    s.train()
    # 4 loader batches / accum 2 -> 2 optimizer steps per epoch
    assert s.iter_count == 2
    assert s.total_iters == 2
    # and the accumulated path still trains: compare one manual pair
    g = torch.Generator().manual_seed(6)
    x = torch.rand(2, 3, 48, 48, generator=g)
    t = (torch.rand(2, 1, 48, 48, generator=g) > 0.5).float()
    before = s.iter_count
    l1 = s.step(x, t)          # micro-batch: no optimizer step
    assert s.iter_count == before
    l2 = s.step(x, t)          # final: optimizer steps
    assert s.iter_count == before + 1
    assert torch.isfinite(l1) and torch.isfinite(l2)
