"""Localize the SSIM fwd mismatch (run from repo root or anywhere)."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

from sodkit import _C
from sodkit.ops import reference

torch.manual_seed(5)
z0 = (torch.randn(2, 1, 40, 40, device="cuda") * 2).to(torch.bfloat16)
t = (torch.rand(2, 1, 40, 40, device="cuda") > 0.5).float()

z = z0.contiguous(memory_format=torch.channels_last)
tc = t.contiguous(memory_format=torch.channels_last)

vals = []
for rep in range(4):
    bce, img, ssim_sum, p, stats = _C.loss_fwd(z, tc, True)
    vals.append(ssim_sum.item())
print("gpu ssim_sum repeats:", vals)

# fp32 reference maps on the same p
pf = p  # fp32 sigmoid from the kernel
sm_ref = reference.ssim_map(pf, t)
print("ref ssim_sum:", sm_ref.sum().item())

# rebuild the gpu per-pixel ssim from the stats maps
N, _, H, W = p.shape
total = N * H * W
st = stats.view(5, N, H, W)
mu_p, mu_t, s_pp, s_tt, s_pt = st[0], st[1], st[2], st[3], st[4]
C1, C2 = 0.01 ** 2, 0.03 ** 2
num = (2 * mu_p * mu_t + C1) * (2 * s_pt + C2)
den = (mu_p ** 2 + mu_t ** 2 + C1) * (s_pp + s_tt + C2)
gpu_map = num / den
print("gpu map sum (recomputed):", gpu_map.sum().item())

d = (gpu_map.unsqueeze(1) - sm_ref).abs()
print("per-pixel |diff| max:", d.max().item(), "mean:", d.mean().item())
idx = d.view(-1).argmax().item()
n = idx // (H * W); rem = idx % (H * W)
print("worst pixel: n=%d h=%d w=%d gpu=%f ref=%f" %
      (n, rem // W, rem % W, gpu_map.view(-1)[idx].item(),
       sm_ref.reshape(-1)[idx].item()))
# check reference's own window stats at that pixel vs gpu stats
for name, g, r in [("mu_p", mu_p, None)]:
    pass
# dump a few edge vs interior diffs
print("edge row diff max:", d[:, :, :1, :].max().item(),
      "interior diff max:", d[:, :, 6:-6, 6:-6].max().item())
