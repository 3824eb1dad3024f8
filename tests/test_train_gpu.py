"""GPU end-to-end coverage for every model family / dtype config
(BASELINE.json acceptance ladder)."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def _mk_solver(preset, **kw):
    from sodkit.config import make_config
    from sodkit.engine import Solver
    kw.setdefault("num_workers", 0)
    cfg = make_config(preset, **kw)
    return Solver(cfg)


def _batch(solver, n=2, size=96, with_depth=False, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = solver._to_device(torch.rand(n, 3, size, size, generator=g), cast=True)
    t = solver._to_device((torch.rand(n, 1, size, size, generator=g) > 0.5).float(),
                          cast=False)
    d = (solver._to_device(torch.rand(n, 1, size, size, generator=g), cast=True)
         if with_depth else None)
    return x, t, d


def test_res2net101_fp16_step():
    s = _mk_solver("minet_res2net101_512", batch_size=2, size=96,
                   output_root="/tmp/sodkit_r2n")
    x, t, _ = _batch(s)
    s.total_iters = 10**9
    first = s.step(x, t).item()
    for _ in range(9):
        last = s.step(x, t).item()
    assert last == last and last < first * 1.5, (first, last)


def test_fp16_device_scaler_skips_on_overflow():
    """Sync-free fp16 path: an inf gradient must skip the fused-SGD update
    via the device multiplier (no host round-trip) and halve the scale."""
    s = _mk_solver("minet_res2net101_512", batch_size=2, size=96,
                   output_root="/tmp/sodkit_devscal")
    assert s._use_device_scaler and s.dynamic_scale
    x, t, _ = _batch(s)
    s.total_iters = 10**9
    s.step(x, t)
    scale0, _ = s._scaler_state()
    before = [p.detach().clone() for p in s.model.parameters()]
    orig_finalize = s.reducer.finalize

    def poisoned():
        orig_finalize()
        s.reducer.buckets[0].flat.view(-1)[0] = float("inf")

    s.reducer.finalize = poisoned
    s.step(x, t)
    s.reducer.finalize = orig_finalize
    scale1, skips = s._scaler_state()
    assert scale1 == scale0 / 2 and skips == 1
    for p, b in zip(s.model.parameters(), before):
        assert torch.equal(p.detach(), b)
    s.step(x, t)  # recovers: params move again
    moved = any(not torch.equal(p.detach(), b)
                for p, b in zip(s.model.parameters(), before))
    assert moved


def test_minet_vgg16_bf16_step():
    s = _mk_solver("minet_r50_352", model="minet_vgg16", batch_size=2, size=96,
                   output_root="/tmp/sodkit_vgg")
    x, t, _ = _batch(s)
    loss = s.step(x, t).item()
    assert loss == loss


def test_eval_and_checkpoint_on_gpu(tmp_path):
    from sodkit.engine.checkpoint import exp_paths, load_checkpoint
    s = _mk_solver("minet_r50_352", batch_size=2, size=96, val_samples=4,
                   output_root=str(tmp_path), epoch_num=1, train_samples=4,
                   log_interval=1)
    s.train()
    res = s.test(save_png=True, max_png=2)
    assert 0.0 <= res["mae"] <= 1.0
    assert 0.0 <= res["maxf"] <= 1.0
    p = exp_paths(str(tmp_path), s.cfg.exp_name)
    assert os.path.exists(p["checkpoint_final"])
    # resume restores params bitwise
    s2 = _mk_solver("minet_r50_352", batch_size=2, size=96,
                    output_root=str(tmp_path))
    load_checkpoint(p, s2.model, s2.optimizer, map_location=s2.device)
    for a, b in zip(s.model.parameters(), s2.model.parameters()):
        assert torch.equal(a, b)


def test_hdfnet_vgg16_step():
    s = _mk_solver("hdfnet_r50_352_ddp", model="hdfnet_vgg16", batch_size=2,
                   size=96, output_root="/tmp/sodkit_hvgg")
    x, t, d = _batch(s, with_depth=True)
    loss = s.step(x, t, d).item()
    assert loss == loss and loss < 100


def test_bf16_vs_fp32_reference_one_step():
    """Full-model forward loss parity: bf16 sodkit model vs the same weights
    run through the fp32 reference ops on CPU."""
    from sodkit.models import build_model
    from sodkit.ops import hybrid_loss
    torch.manual_seed(0)
    m = build_model("minet_vgg16")
    g = torch.Generator().manual_seed(1)
    x = torch.rand(2, 3, 64, 64, generator=g)
    t = (torch.rand(2, 1, 64, 64, generator=g) > 0.5).float()
    m.eval()
    with torch.no_grad():
        ref = hybrid_loss(m(x), t).item()
        mg = build_model("minet_vgg16")
        mg.load_state_dict(m.state_dict())
        mg = mg.cuda()
        from sodkit.models import convert_precision
        convert_precision(mg, torch.bfloat16)
        mg.eval()
        xg = x.cuda().to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
        got = hybrid_loss(mg(xg), t.cuda()).item()
    assert abs(got - ref) < 0.05 * abs(ref) + 0.05, (got, ref)


def test_grad_accumulation_on_gpu(tmp_path):
    """accum_steps=2: the no_sync micro-batch path runs the same HIP
    kernels; two micro-batches then one fused-SGD step, loss finite and
    params actually move."""
    s = _mk_solver("minet_r50_352", batch_size=2, size=96,
                   output_root=str(tmp_path), accum_steps=2,
                   exp_name="accum_gpu")
    s.total_iters = 100
    x, t, _ = _batch(s, n=2, size=96)
    p0 = s.model.classifier.weight.detach().clone()
    l1 = s.step(x, t)                      # micro-batch: no optimizer step
    assert s.iter_count == 0
    assert torch.equal(p0, s.model.classifier.weight.detach())
    l2 = s.step(x, t)                      # final: SGD runs
    assert s.iter_count == 1
    assert torch.isfinite(l1) and torch.isfinite(l2)
    assert not torch.equal(p0, s.model.classifier.weight.detach())


def test_hipgraph_step_matches_eager(tmp_path):
    """cfg.hip_graph: the captured whole-step replay must reproduce the
    eager trajectory (same seeds/data) — graphs change dispatch, not math."""
    kw = dict(batch_size=2, size=96, lr=0.01, output_root=str(tmp_path))
    se = _mk_solver("minet_r50_352", exp_name="eager", **kw)
    sg = _mk_solver("minet_r50_352", exp_name="graph", hip_graph=True, **kw)
    assert sg._graph_ok
    for a, b in zip(se.model.parameters(), sg.model.parameters()):
        assert torch.equal(a, b)   # same seed -> identical init
    se.total_iters = sg.total_iters = 100
    x, t, _ = _batch(se)
    le = lg = None
    for i in range(6):
        le = se.step(x, t).item()
        lg = sg.step(x, t).item()
    assert sg._graph is not None, "graph never captured"
    assert lg == lg and abs(le - lg) < 0.02 * abs(le) + 1e-3, (le, lg)
    for a, b in zip(se.model.parameters(), sg.model.parameters()):
        assert torch.allclose(a.float(), b.float(), atol=1e-2, rtol=1e-2)


def test_hipgraph_full_train_epoch(tmp_path):
    """Graphs must compose with the real epoch loop: dataloader batches,
    the eager TB-panel forward between replays, checkpointing and eval."""
    s = _mk_solver("minet_r50_352", batch_size=2, size=96, val_samples=4,
                   train_samples=8, epoch_num=2, log_interval=2,
                   hip_graph=True, output_root=str(tmp_path),
                   exp_name="graph_train")
    s.train()
    assert s._graph is not None, "graph never captured during train()"
    res = s.test(save_png=False)
    assert 0.0 <= res["mae"] <= 1.0


def test_minet_step_with_8p_backend(tmp_path):
    """Full MINet-R50 step with the opt-in 8-phase conv backend enabled:
    eligible layers (Cout%256 fwd / Cin%256 dgrad) route to
    conv_igemm8.hip, the rest stay on the 128-square kernel — the mixed
    dispatch must train."""
    from sodkit.ops import conv as conv_mod
    s = _mk_solver("minet_r50_352", batch_size=2, size=96,
                   output_root=str(tmp_path), exp_name="ig8")
    s.total_iters = 100
    x, t, _ = _batch(s, n=2, size=96)
    l0 = s.step(x, t).item()
    prev = conv_mod._8P_ENV
    conv_mod._8P_ENV = "1"
    try:
        l1 = s.step(x, t).item()
        l2 = s.step(x, t).item()
    finally:
        conv_mod._8P_ENV = prev
    assert l0 == l0 and l1 == l1 and l2 == l2, "NaN loss under 8p"
    assert l2 < l0 * 1.5, (l0, l1, l2)   # still optimizing, not diverging


def test_hipgraph_with_loader_workers(tmp_path):
    """Regression: under hipGraph replay the host free-runs, and the
    loader's pinned staging buffers were recycled before the async H2D
    copies executed — replays trained on garbage (constant loss, random
    eval).  The one-step-ahead fence + batch holds must keep real-loader
    graph training learning."""
    s = _mk_solver("minet_r50_352", batch_size=4, size=96, train_samples=96,
                   val_samples=8, epoch_num=4, output_root=str(tmp_path),
                   hip_graph=True, num_workers=2, log_interval=8)
    assert s._graph_ok
    s.train()
    assert s._graph is not None, "graph capture must have engaged"
    res = s.test()
    assert res["mae"] < 0.35, f"graph-mode loader training did not learn: {res}"
