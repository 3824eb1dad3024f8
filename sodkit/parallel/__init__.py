"""Distributed bring-up + the hand-built bucketed gradient engine (C1-C3)."""
from .dist import (cleanup, get_local_rank, get_rank, get_world_size,  # noqa: F401
                   is_main, setup)
from .reducer import GradReducer  # noqa: F401
