import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from sodkit import _C

torch.manual_seed(0)
CL = torch.channels_last
for (C, nsplit, H, st, dt) in [(224, 4, 128, 2, torch.float16),
                               (224, 4, 128, 1, torch.float16),
                               (224, 4, 128, 2, torch.bfloat16),
                               (96, 3, 20, 1, torch.bfloat16)]:
    w = (torch.randn(56, 3, 3, C // nsplit, device="cuda") * 0.1).to(dt)
    xfull = torch.randn(4, C, H, H, device="cuda").to(dt).contiguous(memory_format=CL)
    width = C // nsplit
    for i in range(nsplit):
        sl = xfull[:, i * width:(i + 1) * width]
        y_v = _C.conv2d_fwd(sl, w, torch.Tensor(), st, 1, 1, C)
        y_c = _C.conv2d_fwd(sl.contiguous(memory_format=CL), w, torch.Tensor(),
                            st, 1, 1, width)
        nan_v = (~torch.isfinite(y_v.float())).sum().item()
        nan_c = (~torch.isfinite(y_c.float())).sum().item()
        diff = (y_v.float() - y_c.float()).abs().max().item() if nan_v == 0 else -1
        print(f"C={C} split={i} H={H} st={st} {dt}: nan_view={nan_v} "
              f"nan_contig={nan_c} maxdiff={diff}")
