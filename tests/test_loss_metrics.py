import numpy as np
import torch

from sodkit.metrics import MAEMeter, PRFMeter, cal_mae, cal_pr_maxf
from sodkit.ops import reference


def test_hybrid_loss_gradcheck_fp64():
    torch.manual_seed(0)
    z = torch.randn(2, 1, 16, 16, dtype=torch.float64, requires_grad=True)
    t = (torch.rand(2, 1, 16, 16, dtype=torch.float64) > 0.5).double()

    def f(z_):
        return reference.hybrid_loss(z_, t, 1.0, 1.0, 1.0)

    assert torch.autograd.gradcheck(f, (z,), eps=1e-6, atol=1e-4)


def test_hybrid_loss_perfect_prediction_is_low():
    t = (torch.rand(1, 1, 32, 32) > 0.5).float()
    z_good = (t * 2 - 1) * 12.0   # confident correct logits
    z_bad = -(t * 2 - 1) * 12.0
    lg = reference.hybrid_loss(z_good, t)
    lb = reference.hybrid_loss(z_bad, t)
    assert lg.item() < 0.2 and lb.item() > 2.0


def test_ssim_identity_is_one():
    p = torch.rand(1, 1, 24, 24)
    s = reference.ssim_map(p, p)
    assert float(s.min()) > 0.999


def test_cal_mae():
    p = np.zeros((8, 8), np.float32)
    m = np.ones((8, 8), np.float32)
    assert cal_mae(p, m) == 1.0
    assert cal_mae(m, m) == 0.0


def test_prf_perfect_prediction():
    pred = np.zeros((16, 16), np.float32)
    pred[4:12, 4:12] = 1.0
    mask = pred.copy()
    _, _, maxf, _ = cal_pr_maxf(pred, mask)
    assert maxf > 0.99


def test_meters_match_per_image_fns():
    torch.manual_seed(1)
    pred = torch.rand(3, 1, 16, 16)
    mask = (torch.rand(3, 1, 16, 16) > 0.5).float()
    mm = MAEMeter()
    mm.update(pred, mask)
    want = np.mean([cal_mae(pred[i, 0].numpy(), mask[i, 0].numpy())
                    for i in range(3)])
    assert abs(mm.compute() - want) < 1e-6
    pf = PRFMeter()
    pf.update(pred, mask)
    out = pf.compute()
    assert 0 <= out["maxf"] <= 1
    # macro-average protocol: the meter's curve is the MEAN of per-image
    # P/R curves (reference utils/metric.py protocol), not pooled TP/FP
    ps, rs = [], []
    for i in range(3):
        p_i, r_i, _, _ = cal_pr_maxf(pred[i, 0].numpy(), mask[i, 0].numpy())
        ps.append(p_i)
        rs.append(r_i)
    assert np.allclose(out["precision"], np.mean(ps, axis=0), atol=1e-9)
    assert np.allclose(out["recall"], np.mean(rs, axis=0), atol=1e-9)


def test_cel_and_hel_losses():
    from sodkit.losses import CELLoss, HELLoss
    torch.manual_seed(3)
    z = torch.randn(2, 1, 24, 24)
    t = (torch.rand(2, 1, 24, 24) > 0.5).float()
    for crit in (CELLoss(), HELLoss()):
        z2 = z.clone().requires_grad_()
        loss = crit(z2, t)
        loss.backward()
        assert loss.item() == loss.item() and z2.grad is not None
    # CEL is ~0 on a perfect confident prediction
    zgood = (t * 2 - 1) * 12.0
    assert CELLoss(w_bce=0.0)(zgood, t).item() < 0.01
    assert HELLoss()(zgood, t).item() < 0.02


def test_build_loss_dispatch():
    from sodkit.config import make_config
    from sodkit.losses import build_loss, HybridLoss, CELLoss, HELLoss
    assert isinstance(build_loss(make_config("cpu_plumbing")), HybridLoss)
    assert isinstance(build_loss(make_config("cpu_plumbing", loss="cel")), CELLoss)
    assert isinstance(build_loss(make_config("cpu_plumbing", loss="hel")), HELLoss)
