import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from sodkit import _C

d = torch.load("/tmp/nan_case.pt") if os.path.exists("/tmp/nan_case.pt") else None
if d is None:
    # regenerate the capture (deterministic: seed 3407, gen 0, 10 steps)
    from sodkit.config import make_config
    from sodkit.engine import Solver
    cfg = make_config("minet_res2net101_512", batch_size=16, num_workers=0,
                      output_root="/tmp/cap0", log_interval=1000, seed=3407)
    s = Solver(cfg)
    g = torch.Generator().manual_seed(0)
    for i in range(10):
        x = s._to_device(torch.rand(16, 3, 512, 512, generator=g), cast=True)
        t = s._to_device((torch.rand(16, 1, 512, 512, generator=g) > 0.5).float(), cast=False)
        s.step(x, t)
    s.model.eval()
    mod = dict(s.model.named_modules())["backbone.layer2.1.conv1.conv"]
    cap = {}
    mod.register_forward_pre_hook(lambda m, inp: cap.update(x=inp[0]))
    with torch.no_grad():
        s.model(x)
    d = {"x": cap["x"], "w": mod.weight, "stride": 1, "pad": 0, "dil": 1, "ld": cap["x"].shape[1]}

x, w = d["x"].cuda(), d["w"].cuda()
y = _C.conv2d_fwd(x, w, torch.Tensor(), d["stride"], d["pad"], d["dil"], d["ld"])
bad = ~torch.isfinite(y.float())
print("nan:", bad.sum().item(), "of", y.numel())
idx = bad.nonzero()
print("n range:", idx[:,0].min().item(), idx[:,0].max().item())
print("c range:", idx[:,1].min().item(), idx[:,1].max().item(),
      "c histogram buckets:", torch.histc(idx[:,1].float(), 8, 0, 224).tolist())
print("h range:", idx[:,2].min().item(), idx[:,2].max().item())
print("sample coords:", idx[:5].tolist())
# recompute 3 bad outputs in fp64
for j in range(3):
    n, c, h, ww = idx[j].tolist()
    xin = x[n, :, h, ww].double()
    wv = w[c, 0, 0, :].double()
    dot = (xin * wv).sum().item()
    print(f"  ({n},{c},{h},{ww}): fp64 dot = {dot:.4e}, kernel = {y[n,c,h,ww].item()}")
    print("    x absmax:", xin.abs().max().item(), "num@65504:", (xin.abs() == 65504).sum().item())
# also check: does the REG (non-glds) path reproduce?  force via odd ld trick not possible;
# try single-buffer env
os.environ["SODKIT_IGEMM_SB"] = "1"
y2 = _C.conv2d_fwd(x, w, torch.Tensor(), 1, 0, 1, d["ld"])
print("single-buffer nan:", (~torch.isfinite(y2.float())).sum().item())
