"""Shared test config: registers the `gpu` marker (auto-skipped off-GPU).

The suite is designed for the driver's SERIAL runs (`-x -q -m "not gpu"`);
it also passes under pytest-xdist, but the gloo spawn tests want a couple
of cores each (worker intraop threads are pinned to 2 in test_parallel_gloo
for exactly that reason)."""

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a ROCm GPU (run on the MI355X box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
