"""Distributed bring-up — SURVEY §3.4: torchrun-compatible env contract
(RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT), set_device
before first allocation, init_process_group before model wrap.  Backend
"nccl" IS RCCL on PyTorch-ROCm; gloo serves the CPU paths."""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def get_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def get_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


def get_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def is_main() -> bool:
    return get_rank() == 0


def setup(backend: str = "auto", device: str = "auto") -> torch.device:
    """Initialize the process group (if WORLD_SIZE > 1 or env is set) and
    return this rank's device.  Safe to call in a single process with no
    torchrun env: skips init and returns the device."""
    use_cuda = torch.cuda.is_available() and device != "cpu"
    if use_cuda:
        # modulo lets >1 rank share a GPU (CPU-backend integration smokes);
        # on a real node local_rank < device_count and this is the identity
        idx = get_local_rank() % max(torch.cuda.device_count(), 1)
        torch.cuda.set_device(idx)
        dev = torch.device("cuda", idx)
    else:
        dev = torch.device("cpu")

    want_dist = get_world_size() > 1 or "MASTER_ADDR" in os.environ
    if want_dist and not dist.is_initialized():
        if backend == "auto":
            backend = "nccl" if use_cuda else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        dist.init_process_group(backend=backend, rank=get_rank(),
                                world_size=get_world_size(),
                                timeout=datetime.timedelta(minutes=10))
    return dev


def cleanup() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()


def barrier() -> None:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.barrier()
