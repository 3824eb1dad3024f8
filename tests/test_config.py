from sodkit.config import Config, PRESETS, make_config
from sodkit.cli import parse_config


def test_presets_cover_baseline_configs():
    # the five BASELINE.json acceptance configs
    assert set(PRESETS) == {"cpu_plumbing", "minet_r50_352", "minet_r50_352_ddp",
                            "hdfnet_r50_352_ddp", "minet_res2net101_512"}
    assert PRESETS["minet_r50_352"]["dtype"] == "bf16"
    assert PRESETS["minet_res2net101_512"]["dtype"] == "fp16"
    assert PRESETS["minet_res2net101_512"]["size"] == 512


def test_json_roundtrip():
    cfg = make_config("minet_r50_352", lr=0.123)
    cfg2 = Config.from_json(cfg.to_json())
    assert cfg2 == cfg


def test_cli_overrides():
    cfg = parse_config(["--preset", "cpu_plumbing", "--batch_size", "3",
                        "--lr", "0.5", "--test_only", "true"])
    assert cfg.batch_size == 3 and cfg.lr == 0.5 and cfg.test_only
    assert cfg.model == "minet_vgg16"


def test_cli_bool_flag_styles():
    from sodkit.cli import parse_config
    # value style
    c = parse_config(["--preset", "cpu_plumbing", "--resume", "true"])
    assert c.resume is True
    c = parse_config(["--preset", "cpu_plumbing", "--resume", "false"])
    assert c.resume is False
    # bare-flag style (the docstring's `--test_only --resume`)
    c = parse_config(["--preset", "cpu_plumbing", "--test_only", "--resume"])
    assert c.test_only is True and c.resume is True


def test_cli_test_only_resume_requires_checkpoint(tmp_path):
    from sodkit.cli import main
    import pytest as _pytest
    with _pytest.raises(FileNotFoundError):
        main(["--preset", "cpu_plumbing", "--output_root", str(tmp_path),
              "--test_only", "--resume", "--size", "48",
              "--val_samples", "2", "--num_workers", "0"])


def test_cli_test_only_resume_restores_weights(tmp_path):
    from sodkit.cli import main
    import torch
    from sodkit.engine.checkpoint import exp_paths
    # train one tiny epoch, then eval-only with resume must succeed
    args = ["--preset", "cpu_plumbing", "--output_root", str(tmp_path),
            "--size", "48", "--train_samples", "4", "--val_samples", "2",
            "--epoch_num", "1", "--num_workers", "0"]
    assert main(args) == 0
    p = exp_paths(str(tmp_path), "cpu_plumbing")
    import os
    assert os.path.exists(p["checkpoint_final"])
    assert main(args + ["--test_only", "--resume"]) == 0
