"""The CPU reference ops (sodkit/ops/reference.py) are the oracle every GPU
numerics test compares against — pin them directly to the equivalent torch
built-ins so the oracle itself is anchored."""

import torch
import torch.nn.functional as F

from sodkit.ops import reference as R


def test_conv2d_matches_torch():
    g = torch.Generator().manual_seed(0)
    x = torch.randn(2, 5, 9, 9, generator=g, dtype=torch.float64)
    w_krsc = torch.randn(7, 3, 3, 5, generator=g, dtype=torch.float64)
    b = torch.randn(7, generator=g, dtype=torch.float64)
    for stride, pad, dil in [(1, 1, 1), (2, 1, 1), (1, 2, 2), (1, 0, 1)]:
        y = R.conv2d(x, w_krsc, b, stride, pad, dil)
        ref = F.conv2d(x, w_krsc.permute(0, 3, 1, 2), b, stride, pad, dil)
        assert torch.allclose(y, ref, atol=1e-12), (stride, pad, dil)


def test_bn_act_training_matches_torch():
    g = torch.Generator().manual_seed(1)
    x = (torch.randn(3, 6, 7, 7, generator=g) * 2 + 0.5).requires_grad_(True)
    gamma = torch.rand(6, generator=g).requires_grad_(True)
    beta = torch.randn(6, generator=g).requires_grad_(True)
    rm, rv = torch.zeros(6), torch.ones(6)
    x2 = x.detach().clone().requires_grad_(True)
    g2 = gamma.detach().clone().requires_grad_(True)
    b2 = beta.detach().clone().requires_grad_(True)
    rm2, rv2 = rm.clone(), rv.clone()

    y = R.bn_act(x, gamma, beta, rm, rv, training=True, momentum=0.1,
                 eps=1e-5, relu=True, residual=None, process_group=None)
    ref = F.relu(F.batch_norm(x2, rm2, rv2, g2, b2, training=True,
                              momentum=0.1, eps=1e-5))
    assert torch.allclose(y, ref, atol=1e-5)
    assert torch.allclose(rm, rm2, atol=1e-6)   # running stats incl.
    assert torch.allclose(rv, rv2, atol=1e-5)   # unbiased-var convention
    y.square().sum().backward()
    ref.square().sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)
    assert torch.allclose(gamma.grad, g2.grad, atol=1e-4)
    assert torch.allclose(beta.grad, b2.grad, atol=1e-4)


def test_bn_act_eval_matches_torch():
    g = torch.Generator().manual_seed(2)
    x = torch.randn(2, 4, 5, 5, generator=g)
    gamma, beta = torch.rand(4, generator=g), torch.randn(4, generator=g)
    rm = torch.randn(4, generator=g) * 0.1
    rv = torch.rand(4, generator=g) + 0.5
    y = R.bn_act(x, gamma, beta, rm.clone(), rv.clone(), training=False,
                 momentum=0.1, eps=1e-5, relu=False, residual=None,
                 process_group=None)
    ref = F.batch_norm(x, rm, rv, gamma, beta, training=False,
                       momentum=0.1, eps=1e-5)
    assert torch.allclose(y, ref, atol=1e-5)


def test_bilinear2d_matches_torch_interpolate():
    g = torch.Generator().manual_seed(3)
    x = torch.randn(2, 3, 7, 9, generator=g, dtype=torch.float64)
    for oh, ow in [(14, 18), (13, 17), (3, 4), (7, 9)]:
        y = R.bilinear2d(x, oh, ow)
        ref = F.interpolate(x, size=(oh, ow), mode="bilinear",
                            align_corners=False)
        assert torch.allclose(y, ref, atol=1e-12), (oh, ow)


def test_dynamic_filter_matches_unfold():
    g = torch.Generator().manual_seed(4)
    x = torch.randn(2, 8, 6, 6, generator=g, dtype=torch.float64)
    k = torch.randn(2, 9, 6, 6, generator=g, dtype=torch.float64)
    for dil in (1, 2):
        y = R.dynamic_filter(x, k, dil)
        # reference-of-reference: unfold-based per-pixel 3x3 filtering
        cols = F.unfold(x, 3, dilation=dil, padding=dil)          # [N,C*9,HW]
        cols = cols.view(2, 8, 9, 36)
        ref = (cols * k.view(2, 1, 9, 36)).sum(2).view(2, 8, 6, 6)
        assert torch.allclose(y, ref, atol=1e-12), dil


def test_sgd_step_matches_torch_sgd():
    g = torch.Generator().manual_seed(5)
    p_ref = torch.randn(40, generator=g)
    grad = torch.randn(40, generator=g)
    p = p_ref.clone()
    master = p.clone().double().float()
    mom = torch.zeros(40)
    opt = torch.optim.SGD([p_ref.requires_grad_()], lr=0.1, momentum=0.9,
                          weight_decay=1e-4, nesterov=True)
    for _ in range(3):
        p_ref.grad = grad.clone()
        opt.step()
        R.sgd_step(p, grad.clone(), master, mom, lr=0.1, momentum=0.9,
                   weight_decay=1e-4, nesterov=True, grad_scale=1.0)
    assert torch.allclose(p, p_ref.detach(), atol=1e-5)
