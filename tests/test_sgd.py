import torch

from sodkit.ops import SodSGD


def test_sodsgd_matches_torch_sgd_fp32():
    torch.manual_seed(0)
    w1 = torch.randn(10, 10)
    ref_p = torch.nn.Parameter(w1.clone())
    my_p = torch.nn.Parameter(w1.clone())
    opt_ref = torch.optim.SGD([ref_p], lr=0.1, momentum=0.9, weight_decay=1e-2)
    opt_my = SodSGD([{"params": [my_p], "weight_decay": 1e-2}], momentum=0.9)
    for step in range(5):
        g = torch.randn(10, 10)
        ref_p.grad = g.clone()
        my_p.grad = g.clone()
        opt_ref.step()
        opt_my.step(lr=0.1)
        assert torch.allclose(ref_p, my_p, atol=1e-6), step


def test_sodsgd_grad_scale_folds_average():
    torch.manual_seed(0)
    p1 = torch.nn.Parameter(torch.ones(4))
    p2 = torch.nn.Parameter(torch.ones(4))
    o1 = SodSGD([{"params": [p1], "weight_decay": 0.0}], momentum=0.0)
    o2 = SodSGD([{"params": [p2], "weight_decay": 0.0}], momentum=0.0)
    p1.grad = torch.full((4,), 8.0)
    p2.grad = torch.full((4,), 1.0)
    o1.step(lr=0.1, grad_scale=1 / 8)
    o2.step(lr=0.1)
    assert torch.allclose(p1, p2)


def test_sodsgd_state_roundtrip():
    p = torch.nn.Parameter(torch.randn(6))
    opt = SodSGD([{"params": [p], "weight_decay": 0.0}], momentum=0.9)
    p.grad = torch.randn(6)
    opt.step(lr=0.1)
    sd = {k: (v if not isinstance(v, list) else v) for k, v in opt.state_dict().items()}
    p2 = torch.nn.Parameter(p.detach().clone())
    opt2 = SodSGD([{"params": [p2], "weight_decay": 0.0}], momentum=0.9)
    opt2.load_state_dict(sd)
    g = torch.randn(6)
    p.grad = g.clone()
    p2.grad = g.clone()
    opt.step(lr=0.05)
    opt2.step(lr=0.05)
    assert torch.allclose(p, p2)
