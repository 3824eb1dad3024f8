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
    next_epoch = load_checkpoint(p, s2.model, s2.optimizer)
    assert next_epoch == 1
    # params and optimizer masters bitwise equal after resume
    for a, b in zip(s.model.parameters(), s2.model.parameters()):
        assert torch.equal(a, b)
    for pa, pb in zip(s.optimizer.params(), s2.optimizer.params()):
        assert torch.equal(s.optimizer.state[pa]["momentum_buffer"],
                           s2.optimizer.state[pb]["momentum_buffer"])
        assert torch.equal(s.optimizer.state[pa]["master"],
                           s2.optimizer.state[pb]["master"])
