import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from sodkit import _C
from sodkit.config import make_config
from sodkit.engine import Solver

for attempt in range(6):
    cfg = make_config("minet_res2net101_512", batch_size=16, num_workers=0,
                      output_root=f"/tmp/cap{attempt}", log_interval=1000,
                      seed=3407 + attempt)
    s = Solver(cfg)
    g = torch.Generator().manual_seed(attempt)
    for i in range(10):
        x = s._to_device(torch.rand(16, 3, 512, 512, generator=g), cast=True)
        t = s._to_device((torch.rand(16, 1, 512, 512, generator=g) > 0.5).float(), cast=False)
        s.step(x, t)
    s.model.eval()
    caps = {}
    mods = dict(s.model.named_modules())
    from sodkit.models.blocks import SodConv2d
    for n, m in mods.items():
        if isinstance(m, SodConv2d):
            m.register_forward_pre_hook(
                (lambda nm: lambda mod, inp: caps.setdefault(nm, None) or
                    caps.__setitem__(nm, (inp[0], mod)))(n))
    with torch.no_grad():
        s.model(x)
    found = False
    for n, (xin, mod) in caps.items():
        if xin is None or not isinstance(xin, torch.Tensor):
            continue
        ld = (xin.stride(3) if xin.stride(1) == 1 and not xin.is_contiguous(
            memory_format=torch.channels_last) else xin.shape[1])
        try:
            y = _C.conv2d_fwd(xin, mod.weight, torch.Tensor(), mod.stride,
                              mod.padding, mod.dilation, ld)
        except Exception:
            continue
        nan = (~torch.isfinite(y.float())).sum().item()
        fin = torch.isfinite(xin.float()).all().item() and \
              torch.isfinite(mod.weight.float()).all().item()
        if nan and fin:
            print(f"attempt {attempt}: CAPTURED {n} nan={nan} shape={tuple(xin.shape)} ld={ld} st={mod.stride} pad={mod.padding} dil={mod.dilation}")
            torch.save({"x": xin.cpu(), "w": mod.weight.cpu(), "stride": mod.stride,
                        "pad": mod.padding, "dil": mod.dilation, "ld": ld,
                        "y": y.cpu()}, "gpurun_out/nan_case.pt")
            found = True
            break
    if found:
        break
    print(f"attempt {attempt}: clean")
