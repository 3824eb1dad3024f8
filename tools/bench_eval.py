"""Inference throughput: eval-mode forward + sigmoid on pre-staged batches
(the serving-shaped number next to bench.py's training imgs/s).

    python tools/bench_eval.py [model] [size] [batch] [iters]
"""

import sys
import time

import torch

sys.path.insert(0, ".")
from sodkit.models import build_model, convert_precision  # noqa: E402

model_name = sys.argv[1] if len(sys.argv) > 1 else "minet_res50"
size = int(sys.argv[2]) if len(sys.argv) > 2 else 352
batch = int(sys.argv[3]) if len(sys.argv) > 3 else 32
iters = int(sys.argv[4]) if len(sys.argv) > 4 else 30

dev = torch.device("cuda:0")
torch.manual_seed(0)
m = build_model(model_name).to(dev)
convert_precision(m, torch.bfloat16)
m.eval()

with_depth = model_name.startswith("hdfnet")
g = torch.Generator(device="cpu").manual_seed(7)
img = torch.rand(batch, 3, size, size, generator=g).to(dev, torch.bfloat16)
img = img.contiguous(memory_format=torch.channels_last)
dep = None
if with_depth:
    dep = torch.rand(batch, 1, size, size, generator=g).to(dev, torch.bfloat16)
    dep = dep.contiguous(memory_format=torch.channels_last)

with torch.no_grad():
    for _ in range(5):   # warmup
        out = m(img, dep) if dep is not None else m(img)
        pred = torch.sigmoid(out.float())
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        out = m(img, dep) if dep is not None else m(img)
        pred = torch.sigmoid(out.float())
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0

print(f"eval inference {model_name} {size}px bf16 b{batch}: "
      f"{batch * iters / dt:.1f} imgs/s ({dt / iters * 1e3:.2f} ms/batch)")
