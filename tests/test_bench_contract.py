"""bench.py is the driver contract — validate its JSON line end-to-end on
CPU (tiny config; the driver runs the real thing on MI355X)."""

import json
import subprocess
import sys
import os

import pytest


@pytest.mark.timeout(600)
def test_bench_json_contract(tmp_path):
    env = dict(os.environ)
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(root, "bench.py"),
         "--model", "minet_vgg16", "--size", "64", "--batch", "2",
         "--dtype", "fp32", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=500, env=env, cwd=root)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    j = json.loads(line)                      # exactly one JSON line
    assert j["metric"] == "imgs_per_sec"
    assert j["unit"] == "imgs/s"
    assert j["n_gpus"] == 1 and j["steps"] == 2 and j["warmup"] == 1
    assert j["higher_is_better"] is True
    assert j["scaling"] == "weak"
    assert j["vs_baseline"] is None           # no published baseline exists
    assert j["data"] == "synthetic"
    assert j["value"] > 0 and j["ms_per_step"] > 0
    # value/ms_per_step consistency: value = batch*world*1000/ms
    assert abs(j["value"] - j["config"]["global_batch"] * 1000.0 /
               j["ms_per_step"]) / j["value"] < 1e-6
    cfg = j["config"]
    assert cfg == {"model": "minet_vgg16", "global_batch": 2,
                   "seq_len": 64, "parallelism": "dp1"}
