"""Loader for the in-tree HIP extension ``sodkit._C`` (built by setup.py for
gfx950).  GPU execution REQUIRES the native kernels: there is no silent eager
fallback on a CUDA/HIP device (the CPU path lives in ops.reference and is used
only for CPU tensors)."""

from __future__ import annotations

_ext = None


def ext():
    global _ext
    if _ext is None:
        try:
            from sodkit import _C  # type: ignore
        except ImportError as e:  # pragma: no cover - only on mis-built installs
            raise RuntimeError(
                "sodkit._C HIP extension is not built. Build it in-tree with "
                "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
                "sodkit refuses to run GPU tensors through an eager fallback."
            ) from e
        _ext = _C
    return _ext


def has_ext() -> bool:
    try:
        ext()
        return True
    except RuntimeError:
        return False
