"""sodkit — MI355X-native distributed salient-object-detection training framework.

A from-scratch rebuild of the capability surface of
lartpang/Distributed-SOD-Project (the reference mount was empty
at build time, so BASELINE.json is the binding spec): MINet / HDFNet
encoder-decoder SOD networks with ResNet / Res2Net / VGG backbones, trained
with pixel-wise hybrid BCE+IoU+SSIM losses under data-parallel RCCL-over-xGMI
on 8x MI355X.

Compute path: PyTorch-ROCm tensors + hand-written CDNA4 (gfx950) HIP kernels
(`csrc/`, built as `sodkit._C`) for the hot ops — 3x3/1x1 convolutions as
MFMA implicit GEMM with LDS-staged tiles, fused BatchNorm+ReLU, bilinear
resize, the fused hybrid loss, and a fused multi-tensor SGD.  The gradient
engine (`sodkit.parallel.reducer`) is a bucketed, backward-overlapped
all-reduce over RCCL on a side HIP stream.
"""

import torch as _torch  # noqa: F401  (the _C extension links against torch libs)

__version__ = "0.1.0"

from . import config  # noqa: F401
