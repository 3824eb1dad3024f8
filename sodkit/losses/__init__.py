"""Loss modules: fused hybrid BCE+IoU+SSIM, CEL, HEL (SURVEY §2.1 loss/)."""
from .hybrid import HybridLoss  # noqa: F401
from .cel import CELLoss, HELLoss, build_loss  # noqa: F401
