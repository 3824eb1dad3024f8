"""Deferred-grad mode (sodkit/parallel/reducer.py): at world==1 on GPU,
zero_grad() sets .grad=None (autograd assigns fresh grads — no per-param
add_ launches) and finalize() lands them in the bucket flats with one
multicopy kernel.  Must be BIT-identical to the grad-as-view path:
assignment == add-into-zero, and the gather is an exact byte copy."""

import copy
import os

import pytest
import torch

from sodkit.parallel.reducer import GradReducer


def _model(device="cpu"):
    torch.manual_seed(7)
    m = torch.nn.Sequential(
        torch.nn.Conv2d(3, 8, 3, padding=1),
        torch.nn.BatchNorm2d(8),
        torch.nn.ReLU(),
        torch.nn.Conv2d(8, 4, 1),
    )
    return m.to(device)


def _mlp(device):
    # bitwise comparisons need run-to-run DETERMINISTIC backwards: ATen's
    # conv wgrad uses atomics on ROCm, so two separate runs of the same conv
    # model already differ in the last bits.  Linear backward is a plain
    # rocBLAS GEMM (fixed reduction order).
    torch.manual_seed(7)
    m = torch.nn.Sequential(
        torch.nn.Linear(32, 64),
        torch.nn.ReLU(),
        torch.nn.Linear(64, 16),
        torch.nn.ReLU(),
        torch.nn.Linear(16, 4),
    )
    return m.to(device)


def test_defer_off_on_cpu():
    r = GradReducer(_model())
    assert not r.defer  # CPU: gather kernel is GPU-only; views stay


def test_defer_env_kill_switch(monkeypatch):
    monkeypatch.setenv("SODKIT_DEFER_GRADS", "0")
    r = GradReducer(_model())
    assert not r.defer


def _flats_after(model, x, defer: bool, steps: int = 2, accum: int = 1):
    prev = os.environ.get("SODKIT_DEFER_GRADS")
    os.environ["SODKIT_DEFER_GRADS"] = "1" if defer else "0"
    try:
        r = GradReducer(model)
    finally:
        if prev is None:
            os.environ.pop("SODKIT_DEFER_GRADS", None)
        else:
            os.environ["SODKIT_DEFER_GRADS"] = prev
    assert r.defer == (defer and x.is_cuda)
    for _ in range(steps):
        r.zero_grad()
        for _ in range(accum):   # micro-batches accumulate into .grad
            model(x).square().mean().backward()
        r.finalize()
    return [b.flat.clone() for b in r.buckets]


@pytest.mark.gpu
def test_defer_bitwise_vs_views():
    m1 = _mlp("cuda")
    m2 = copy.deepcopy(m1)
    x = torch.randn(8, 32, device="cuda")
    ref = _flats_after(m1, x, defer=False)
    got = _flats_after(m2, x, defer=True)
    assert len(ref) == len(got)
    for a, b in zip(ref, got):
        assert torch.equal(a, b)


@pytest.mark.gpu
def test_defer_bitwise_with_accumulation():
    m1 = _mlp("cuda")
    m2 = copy.deepcopy(m1)
    x = torch.randn(8, 32, device="cuda")
    ref = _flats_after(m1, x, defer=False, steps=1, accum=3)
    got = _flats_after(m2, x, defer=True, steps=1, accum=3)
    for a, b in zip(ref, got):
        assert torch.equal(a, b)


@pytest.mark.gpu
def test_defer_close_on_conv_model():
    # conv backwards are atomically-reduced (nondeterministic last bits), so
    # this one checks tight closeness rather than equality
    m1 = _model("cuda")
    m2 = copy.deepcopy(m1)
    x = torch.randn(2, 3, 16, 16, device="cuda")
    ref = _flats_after(m1, x, defer=False)
    got = _flats_after(m2, x, defer=True)
    for a, b in zip(ref, got):
        assert torch.allclose(a, b, rtol=1e-4, atol=1e-5), \
            (a - b).abs().max().item()


@pytest.mark.gpu
def test_defer_missing_grad_zeroes_slot():
    m = _model("cuda")
    os.environ["SODKIT_DEFER_GRADS"] = "1"
    try:
        r = GradReducer(m)
    finally:
        os.environ.pop("SODKIT_DEFER_GRADS", None)
    assert r.defer
    x = torch.randn(2, 3, 16, 16, device="cuda")
    r.zero_grad()
    m(x).square().mean().backward()
    r.finalize()
    # second step: only the tail conv gets grads -> every other slot must
    # read ZERO (stale grads applied by SGD would corrupt training)
    r.zero_grad()
    feat = m[0](x).detach().requires_grad_(True)
    m[3](m[2](m[1](feat))).square().mean().backward()
    r.finalize()
    zeroed = total = 0
    for p in m[0].parameters():
        total += 1
        if torch.equal(r.grad_views[p].detach(),
                       torch.zeros_like(r.grad_views[p])):
            zeroed += 1
    assert zeroed == total


@pytest.mark.gpu
def test_defer_with_direct_sgd_tables():
    """GradReducer + SodSGD WITHOUT the trainer's grad_src wiring (the
    tools/rccl_validate oracle pattern): the fused SGD bakes .grad pointer
    tables on first step — defer mode must leave .grad pointing at the
    STABLE bucket views after finalize, or the tables bake step-transient
    autograd tensors and later steps read garbage (caught as NaN weights
    in the oracle's 1-rank reference)."""
    from sodkit.ops import SodSGD
    torch.manual_seed(11)
    m = _model("cuda")
    r = GradReducer(m)
    assert r.defer
    opt = SodSGD([{"params": list(m.parameters()), "weight_decay": 1e-4}])
    x = torch.randn(2, 3, 16, 16, device="cuda")
    for _ in range(4):
        r.zero_grad()
        m(x).square().mean().backward()
        r.finalize()
        for p in m.parameters():
            assert p.grad is not None and p.grad.data_ptr() == \
                r.grad_views[p].data_ptr(), "finalize must restore the views"
        opt.step(0.05, r.grad_scale)
    ck = sum(p.detach().float().sum().item() for p in m.parameters())
    assert ck == ck, "weights went NaN: SGD tables read stale grad memory"
