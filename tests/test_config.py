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
