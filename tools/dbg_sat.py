import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from sodkit import _C
CL = torch.channels_last
# acc = 256 * (-273.4) ~ -69990 -> must saturate to -65504, not -inf
x = torch.ones(1, 256, 16, 16, device="cuda", dtype=torch.float16).contiguous(memory_format=CL)
w = torch.full((8, 1, 1, 256), -273.4, device="cuda", dtype=torch.float16)
y = _C.conv2d_fwd(x, w, torch.Tensor(), 1, 0, 1, 256)
print("expect -65504:", y[0, 0, 0, 0].item(), "any inf:", (~torch.isfinite(y.float())).any().item())
# positive side
w2 = torch.full((8, 1, 1, 256), 273.4, device="cuda", dtype=torch.float16)
y2 = _C.conv2d_fwd(x, w2, torch.Tensor(), 1, 0, 1, 256)
print("expect +65504:", y2[0, 0, 0, 0].item())
# with values that are exactly at fp16 max in the INPUT
x3 = torch.full((1, 256, 16, 16), 65504.0, device="cuda", dtype=torch.float16).contiguous(memory_format=CL)
w3 = torch.zeros(8, 1, 1, 256, device="cuda", dtype=torch.float16); w3[:, :, :, 0] = -1.02
y3 = _C.conv2d_fwd(x3, w3, torch.Tensor(), 1, 0, 1, 256)
print("expect -65504 (single-product overflow path):", y3[0, 0, 0, 0].item())
