import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from sodkit.config import make_config
from sodkit.engine import Solver

cfg = make_config("minet_res2net101_512", batch_size=16, num_workers=0,
                  output_root="/tmp/dbgf16", log_interval=100)
s = Solver(cfg)
g = torch.Generator().manual_seed(0)
for i in range(10):
    x = s._to_device(torch.rand(16, 3, 512, 512, generator=g), cast=True)
    t = s._to_device((torch.rand(16, 1, 512, 512, generator=g) > 0.5).float(), cast=False)
    loss = s.step(x, t).item()
print("final train loss", loss, "loss_scale", s.loss_scale)

bad_params = [n for n, p in s.model.named_parameters() if not torch.isfinite(p).all()]
bad_bufs = [n for n, b in s.model.named_buffers() if not torch.isfinite(b).all()]
print("non-finite params:", bad_params[:5], "bufs:", bad_bufs[:5])

s.model.eval()
hits = []
def hook(name):
    def f(mod, inp, out):
        if isinstance(out, torch.Tensor) and not torch.isfinite(out.float()).all():
            if len(hits) < 5:
                frac = (~torch.isfinite(out.float())).float().mean().item()
                hits.append((name, type(mod).__name__, frac))
    return f
for n, m in s.model.named_modules():
    m.register_forward_hook(hook(n))
with torch.no_grad():
    y = s.model(x)
print("first NaN modules:", hits)
print("logits finite:", torch.isfinite(y.float()).all().item())

# isolate the first offending conv with its REAL input
target = "backbone.layer2.0.convs.0.conv"
mod = dict(s.model.named_modules())[target]
captured = {}
def pre(m, inp):
    captured["x"] = inp[0]
mod.register_forward_pre_hook(pre)
with torch.no_grad():
    s.model(x)
xin = captured["x"]
print("captured input:", tuple(xin.shape), xin.dtype, tuple(xin.stride()),
      "finite:", torch.isfinite(xin.float()).all().item(),
      "absmax:", xin.float().abs().max().item())
from sodkit import _C
w = mod.weight
for rep in range(3):
    y = _C.conv2d_fwd(xin, w, torch.Tensor(), mod.stride, mod.padding,
                      mod.dilation, xin.stride(3) if xin.stride(1) == 1 and not xin.is_contiguous(memory_format=torch.channels_last) else xin.shape[1])
    print("rep", rep, "nan:", (~torch.isfinite(y.float())).sum().item())
import torch.nn.functional as F
yr = F.conv2d(xin.float(), w.float().permute(0, 3, 1, 2), None,
              stride=mod.stride, padding=mod.padding, dilation=mod.dilation)
print("ref finite:", torch.isfinite(yr).all().item(), "absmax", yr.abs().max().item())
