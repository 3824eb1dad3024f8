import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch.nn.functional as F
from sodkit import _C
from sodkit.config import make_config
from sodkit.engine import Solver

cfg = make_config("minet_res2net101_512", batch_size=16, num_workers=0,
                  output_root="/tmp/dbgf16b", log_interval=100)
s = Solver(cfg)
g = torch.Generator().manual_seed(0)
for i in range(10):
    x = s._to_device(torch.rand(16, 3, 512, 512, generator=g), cast=True)
    t = s._to_device((torch.rand(16, 1, 512, 512, generator=g) > 0.5).float(), cast=False)
    s.step(x, t)
target = "backbone.layer2.0.convs.0.conv"
mod = dict(s.model.named_modules())[target]
cap = {}
mod.register_forward_pre_hook(lambda m, inp: cap.update(x=inp[0]))
s.model.eval()
with torch.no_grad():
    s.model(x)
xin, w = cap["x"], mod.weight
CL = torch.channels_last

def run(xx, ld=None, tag=""):
    ld = ld if ld is not None else (xx.stride(3) if xx.stride(1) == 1 and
         not xx.is_contiguous(memory_format=CL) else xx.shape[1])
    y = _C.conv2d_fwd(xx, w, torch.Tensor(), 2, 1, 1, ld)
    bad = ~torch.isfinite(y.float())
    n = bad.sum().item()
    if n:
        idx = bad.nonzero()[:4]
        print(f"{tag}: nan={n} first_at={idx.tolist()}")
    else:
        print(f"{tag}: nan=0")
    return y

run(xin, tag="view")
xc = xin.contiguous(memory_format=CL)
run(xc, tag="contig")
run(xc.clamp(-1000, 1000), tag="clamped1k")
xr = torch.randn_like(xc)
xr[0, 0, 0, :] = 65504.0
run(xr, tag="randn+spike")
# how big are the finite outputs around the NaN?
y = run(xin, tag="view2")
print("y absmax finite:", y.float().nan_to_num(0).abs().max().item())
# isolate: bf16 version of the same data
xb = xc.to(torch.bfloat16)
wb = w.to(torch.bfloat16)
yb = _C.conv2d_fwd(xb, wb, torch.Tensor(), 2, 1, 1, xb.shape[1])
print("bf16 same data nan:", (~torch.isfinite(yb.float())).sum().item())
