"""sodkit.ops — the single op-dispatch layer (every hot op goes
through here so the HIP kernels drop in under a stable interface; on GPU the
native path is mandatory and fail-loud, on CPU the torch reference runs)."""

from ._backend import ext, has_ext  # noqa: F401
from .conv import conv2d  # noqa: F401
from .bnrelu import bn_act  # noqa: F401
from .resize import bilinear2d, upsample_add  # noqa: F401
from .loss import hybrid_loss  # noqa: F401
from .pool import maxpool2d_3x3s2  # noqa: F401
from .dynfilter import dynamic_filter  # noqa: F401
from .sgd import SodSGD, split_decay_groups  # noqa: F401
from . import reference  # noqa: F401
