"""Synthetic SOD data (BASELINE.json: replaces the reference dataloader)."""
from .synthetic import SyntheticSOD, create_loader  # noqa: F401
