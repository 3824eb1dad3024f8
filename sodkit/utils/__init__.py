"""Meters, seeding, logging, recorders, PNG writer, profiler glue."""
from .meters import AvgMeter, Timer  # noqa: F401
from .seed import seed_everything  # noqa: F401
from .logging import get_logger, rank0_print  # noqa: F401
