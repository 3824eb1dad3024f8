"""Capability-surface regression guard: every component PARITY.md claims
must exist with the claimed API (SURVEY §2 inventory)."""

import dataclasses
import pathlib
import re


def test_python_surface_complete():
    from sodkit.config import Config, PRESETS                     # noqa: F401
    assert set(PRESETS) == {"cpu_plumbing", "minet_r50_352",
                            "minet_r50_352_ddp", "hdfnet_r50_352_ddp",
                            "minet_res2net101_512"}
    from sodkit.cli import build_argparser, parse_config, main    # noqa: F401
    from sodkit.engine import Solver                              # noqa: F401
    from sodkit.engine.checkpoint import (exp_paths, make_exp_tree,  # noqa
                                          save_checkpoint, load_checkpoint)
    from sodkit.models.blocks import (SodConv2d, SodBatchNorm2d,  # noqa
                                      ConvBNReLU, maxpool2d_3x3s2)
    from sodkit.ops import (conv2d, bn_act, bilinear2d, upsample_add,  # noqa
                            hybrid_loss, dynamic_filter, SodSGD,
                            split_decay_groups)
    from sodkit.losses import build_loss                          # noqa: F401
    from sodkit.losses.cel import CELLoss, HELLoss                # noqa: F401
    from sodkit.metrics import MAEMeter, PRFMeter                 # noqa: F401
    from sodkit.metrics.sod import cal_mae, cal_pr_maxf           # noqa: F401
    from sodkit.parallel import (GradReducer, setup, get_rank,    # noqa
                                 get_world_size)
    from sodkit.data import create_loader, SyntheticSOD           # noqa: F401
    from sodkit.utils import (AvgMeter, Timer, get_logger,        # noqa
                              seed_everything)
    from sodkit.utils.recorder import TBRecorder, ResultsRecorder  # noqa
    from sodkit.utils.png import write_gray_png                   # noqa: F401
    from sodkit.utils.profiler import profile_ctx                 # noqa: F401


def test_model_registry_builds_all_families():
    from sodkit.models import build_model
    for name in ["minet_res50", "minet_vgg16", "minet_res2net101",
                 "hdfnet_res50", "hdfnet_vgg16"]:
        assert build_model(name) is not None


def test_config_covers_reference_flag_surface():
    from sodkit.config import Config
    fields = {f.name for f in dataclasses.fields(Config)}
    needed = {"exp_name", "epoch_num", "batch_size", "lr", "lr_type",
              "lr_power", "warmup_iters", "momentum", "weight_decay",
              "nesterov", "sgd_trick", "dtype", "loss_scale", "resume",
              "test_only", "size", "seed", "data_seed", "sync_bn",
              "bucket_mb", "overlap_comm", "grad_clip", "profile_steps",
              "tb", "loss", "model", "num_workers", "train_samples",
              "val_samples", "output_root", "log_interval",
              "channels_last", "backend", "device"}
    assert needed <= fields, needed - fields


def test_kernel_inventory_sources_present():
    """the hand-written kernel entry points exist in
    csrc (symbol-level; numerics are covered by the @gpu suite)."""
    root = pathlib.Path(__file__).resolve().parents[1] / "csrc"
    src = "".join(p.read_text() for p in root.glob("*.hip")
                  if not p.name.endswith("_hip.hip"))
    for sym in ["igemm_kernel", "wgrad_kernel", "splitk_reduce",
                "bn_stats", "bn_apply", "bn_bwd_reduce", "bn_bwd_apply",
                "bn_reduce_finalize", "slab_reduce", "bilinear_fwd",
                "bilinear_bwd", "loss_pointwise", "ssim_stats", "ssim_coef",
                "loss_bwd", "sgd_fused", "channel_sum"]:
        assert re.search(sym, src), f"kernel symbol missing: {sym}"
    assert re.search(r"dynfilter|dynamic_filter|df_fwd",
                     (root / "dynfilter.hip").read_text())


def test_8p_eligibility_predicate_parity():
    """The python dispatch predicate (sodkit/ops/conv.py) and the C++
    TORCH_CHECK predicate (_C.conv8_eligible) must agree, or an eligible-
    looking layer would raise at runtime on GPU."""
    from sodkit import _C
    for cout in (1, 8, 64, 128, 255, 256, 512, 1024, 2048):
        for cin in (1, 3, 7, 8, 64, 72, 256):
            py = cout % 256 == 0 and cin % 8 == 0     # ops/conv.py fwd gate
            assert _C.conv8_eligible(cout, cin) == py, (cout, cin)


def test_8p_host_guards_without_gpu():
    """The 8p entry points exist in the built extension and their guards
    fire cleanly on CPU tensors (binding arity + TORCH_CHECK path)."""
    import pytest
    import torch
    from sodkit import _C
    x = torch.randn(1, 8, 8, 8)
    w = torch.randn(256, 1, 1, 8)
    with pytest.raises(RuntimeError, match="is_cuda"):
        _C.conv2d_fwd_8p(x, w, torch.Tensor(), 1, 0, 1, 8)
    with pytest.raises(RuntimeError, match="is_cuda"):
        _C.conv2d_dgrad_8p(x, w, 8, 8, 1, 0, 1, 256)


def test_backend_loader_contract():
    """has_ext() finds the in-tree build; ext() returns the module whose
    file lives INSIDE the repo (not site-packages) — the driver's
    native-code check records exactly this."""
    import os
    import sodkit
    from sodkit.ops._backend import ext, has_ext
    assert has_ext()
    mod = ext()
    pkg_dir = os.path.dirname(os.path.abspath(sodkit.__file__))
    assert os.path.dirname(os.path.abspath(mod.__file__)) == pkg_dir
    assert mod.BUILD_ARCH == "gfx950"
