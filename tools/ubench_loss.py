"""Loss kernel microbench: isolate ssim_stats / loss_bwd timing."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from sodkit import _C

N, H, W = 32, 352, 352
z = (torch.randn(N, 1, H, W, device="cuda") * 2).to(torch.bfloat16)
t = (torch.rand(N, 1, H, W, device="cuda") > 0.5).float()
go = torch.ones(1, device="cuda")

def timeit(fn, iters=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters * 1e3

t_fwd_nossim = timeit(lambda: _C.loss_fwd(z, t, False))
t_fwd = timeit(lambda: _C.loss_fwd(z, t, True))
bce, img, ssim_sum, p, stats = _C.loss_fwd(z, t, True)
t_bwd = timeit(lambda: _C.loss_bwd(p, t, img, stats, 1.0, 1.0, 1.0, go))
t_bwd_nossim = timeit(lambda: _C.loss_bwd(p, t, img, torch.empty(0, device="cuda"), 1.0, 1.0, 0.0, go))
print(f"fwd no-ssim {t_fwd_nossim:.2f} ms; fwd {t_fwd:.2f} ms; "
      f"bwd {t_bwd:.2f} ms; bwd no-ssim {t_bwd_nossim:.2f} ms")
