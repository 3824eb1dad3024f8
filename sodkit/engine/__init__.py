"""Trainer/solver loop + reference-layout checkpointing."""
from .trainer import Solver  # noqa: F401
from .checkpoint import load_checkpoint, save_checkpoint  # noqa: F401
