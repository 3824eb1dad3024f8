"""Build the in-tree HIP extension sodkit._C for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands inside sodkit/ so it travels with the repo snapshot to GPU
boxes (no JIT cache dependence)."""

import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))

# torch's hipify step compiles copied csrc/*_hip.hip files whose dep files
# do not track csrc/common.h: a header edit would silently keep stale
# objects.  Invalidate explicitly.
def _invalidate_stale_objects():
    import glob
    hdr = os.path.join(ROOT, "csrc", "common.h")
    if not os.path.exists(hdr):
        return
    hmt = os.path.getmtime(hdr)
    for o in glob.glob(os.path.join(ROOT, "build", "temp*", "csrc", "*.o")):
        if os.path.getmtime(o) < hmt:
            os.remove(o)
    for c in glob.glob(os.path.join(ROOT, "csrc", "*_hip.hip")):
        if os.path.getmtime(c) < hmt:
            os.remove(c)


_invalidate_stale_objects()

SRC = [
    "csrc/bind.cpp",
    "csrc/conv_igemm.hip",
    "csrc/conv_igemm8.hip",
    "csrc/wgrad_dir.hip",
    "csrc/conv_dir.hip",
    "csrc/im2col.hip",
    "csrc/wgrad_k1.hip",
    "csrc/bnrelu.hip",
    "csrc/bn_fused.hip",
    "csrc/bilinear.hip",
    "csrc/loss.hip",
    "csrc/sgd.hip",
    "csrc/multicopy.hip",
    "csrc/maxpool.hip",
    "csrc/dynfilter.hip",
]

setup(
    name="sodkit-kernels",
    ext_modules=[
        CUDAExtension(
            name="sodkit._C",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
