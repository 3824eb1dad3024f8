"""Experiment configuration: one dataclass + named presets.

Capability parity with the reference's ``config.py`` (a flat
experiment dict with exp_name, epoch_num, batch_size, lr, poly LR, sgd_trick,
use_amp, resume, size).  The five BASELINE.json configs are named presets.
Kept deliberately boring: dataclass + argparse overrides, no hydra/yaml.
"""

from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class Config:
    # experiment identity / io
    exp_name: str = "minet_r50_352"
    output_root: str = "output"          # ckpt tree root (reference layout)
    resume: bool = False
    test_only: bool = False

    # model
    model: str = "minet_res50"           # registry key, sodkit.models.build_model
    in_channels: int = 3                 # 4 streams handled by model (hdfnet: rgb+depth)

    # data (synthetic generator — BASELINE.json replaces the DUTS loader)
    size: int = 352                      # square train/test resolution
    train_samples: int = 3200            # synthetic "dataset" length per epoch
    val_samples: int = 128               # fixed-seed synthetic validation split
    batch_size: int = 32                 # per-GPU
    num_workers: int = 8
    data_seed: int = 20260913
    shuffle: bool = True                 # off => sequential sharding, so an
    #                                      N-rank global batch is the SAME
    #                                      sample set as the 1-rank batch
    #                                      (deterministic DDP parity runs)

    # optimization
    epoch_num: int = 3
    lr: float = 0.05
    momentum: float = 0.9
    weight_decay: float = 5e-4
    nesterov: bool = False
    sgd_trick: bool = True               # no weight-decay on bias/BN params
    lr_type: str = "poly"                # poly | const
    lr_power: float = 0.9
    warmup_iters: int = 0
    grad_clip: float = 0.0               # 0 = off

    # hipGraph whole-step capture (single-GPU, accum_steps==1): replaces
    # ~1900 per-step kernel dispatches with one graph launch.  LR feeds the
    # fused SGD through a device tensor so the poly schedule survives replay.
    # Default ON since late r2: real-loader training measured +14% (1148.9
    # vs 1008.9 imgs/s, identical MAE) once capture moved to thread_local
    # error mode and the replay loop fenced the loader's pinned buffers;
    # multi-GPU and accum configs fall back automatically, and a failed
    # capture falls back to eager with a log line.
    hip_graph: bool = True

    # precision
    dtype: str = "bf16"                  # bf16 | fp16 | fp32 (compute dtype on GPU)
    loss_scale: float = 1.0              # static loss scale (fp16 only; bf16 needs none)

    # loss selection + weights (hybrid = the fused BCE+IoU+SSIM kernel;
    # cel / hel are the reference's MINet / HDFNet recipes)
    loss: str = "hybrid"
    w_bce: float = 1.0
    w_iou: float = 1.0
    w_ssim: float = 1.0

    # distributed / engine
    backend: str = "auto"                # auto -> nccl(RCCL) on GPU, gloo on CPU
    bucket_mb: int = 25                  # gradient bucket size for the RCCL reducer
    accum_steps: int = 1                 # gradient-accumulation micro-batches
                                         # per optimizer step (no_sync comm)
    sync_bn: bool = False
    overlap_comm: bool = True            # all-reduce on side stream during backward
    seed: int = 3407

    # logging / obs
    log_interval: int = 10
    tb: bool = False                     # tensorboard scalars (optional dep)
    profile_steps: int = 0               # wrap N steps in torch.profiler if > 0

    # internal / device
    device: str = "auto"                 # auto -> cuda if available else cpu
    channels_last: bool = True           # NHWC activations (the native GPU layout)

    def to_json(self) -> str:
        return json.dumps(dataclasses.asdict(self), indent=2)

    @staticmethod
    def from_json(s: str) -> "Config":
        d = json.loads(s)
        known = {f.name for f in dataclasses.fields(Config)}
        return Config(**{k: v for k, v in d.items() if k in known})

    def replace(self, **kw) -> "Config":
        return dataclasses.replace(self, **kw)


# ---------------------------------------------------------------------------
# The five BASELINE.json acceptance configs (BASELINE.json:7-11), as presets.
# ---------------------------------------------------------------------------

PRESETS: dict[str, dict] = {
    # [BJ:7] MINet-VGG16 128x128 batch=2 on CPU, world_size=1 (plumbing, no GPU)
    "cpu_plumbing": dict(
        exp_name="cpu_plumbing", model="minet_vgg16", size=128, batch_size=2,
        dtype="fp32", epoch_num=1, train_samples=8, val_samples=4,
        num_workers=0, device="cpu", channels_last=False, lr=0.01,
    ),
    # [BJ:8] MINet-ResNet50 352x352 bf16 batch=32 on one MI355X
    "minet_r50_352": dict(
        exp_name="minet_r50_352", model="minet_res50", size=352, batch_size=32,
        dtype="bf16",
    ),
    # [BJ:9] MINet-ResNet50 352x352 DDP on 8x MI355X
    "minet_r50_352_ddp": dict(
        exp_name="minet_r50_352_ddp", model="minet_res50", size=352,
        batch_size=32, dtype="bf16",
    ),
    # [BJ:10] HDFNet-ResNet50 RGB-D 352x352 DDP 8x MI355X
    "hdfnet_r50_352_ddp": dict(
        exp_name="hdfnet_r50_352_ddp", model="hdfnet_res50", size=352,
        batch_size=32, dtype="bf16",
    ),
    # [BJ:11] MINet-Res2Net101 512x512 fp16 DDP 8x MI355X, global batch 512
    "minet_res2net101_512": dict(
        exp_name="minet_res2net101_512", model="minet_res2net101", size=512,
        batch_size=64, dtype="fp16", loss_scale=1024.0,
    ),
}


def make_config(preset: Optional[str] = None, **overrides) -> Config:
    cfg = Config()
    if preset is not None:
        if preset not in PRESETS:
            raise KeyError(f"unknown preset {preset!r}; have {sorted(PRESETS)}")
        cfg = cfg.replace(**PRESETS[preset])
    if overrides:
        cfg = cfg.replace(**overrides)
    return cfg
