from .hybrid import HybridLoss  # noqa: F401
from .cel import CELLoss, HELLoss, build_loss  # noqa: F401
