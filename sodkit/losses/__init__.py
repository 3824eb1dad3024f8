from .hybrid import HybridLoss  # noqa: F401
