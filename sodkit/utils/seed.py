"""Seeding: identical params across ranks at random-init."""

from __future__ import annotations

import os
import random

import numpy as np
import torch


def seed_everything(seed: int) -> None:
    """Seed every RNG identically — must run BEFORE model construction on every
    rank so that random-init weights agree without a broadcast (the init
    broadcast is still done as a belt-and-braces step)."""
    random.seed(seed)
    np.random.seed(seed % (2**32 - 1))
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    os.environ["PYTHONHASHSEED"] = str(seed)
