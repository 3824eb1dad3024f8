"""Rank-0 gated logging to stdout + the reference-layout txt log file."""

from __future__ import annotations

import logging
import os
import sys
from typing import Optional

_loggers: dict[str, logging.Logger] = {}


def _rank() -> int:
    return int(os.environ.get("RANK", "0"))


def get_logger(name: str = "sodkit", log_file: Optional[str] = None) -> logging.Logger:
    key = f"{name}:{log_file}"
    if key in _loggers:
        return _loggers[key]
    logger = logging.getLogger(key)
    logger.setLevel(logging.INFO)
    logger.propagate = False
    fmt = logging.Formatter("[%(asctime)s] %(message)s", datefmt="%H:%M:%S")
    if _rank() == 0:
        sh = logging.StreamHandler(sys.stdout)
        sh.setFormatter(fmt)
        logger.addHandler(sh)
        if log_file is not None:
            os.makedirs(os.path.dirname(log_file), exist_ok=True)
            fh = logging.FileHandler(log_file)
            fh.setFormatter(fmt)
            logger.addHandler(fh)
    else:
        logger.addHandler(logging.NullHandler())
    _loggers[key] = logger
    return logger


def rank0_print(*args, **kw) -> None:
    if _rank() == 0:
        print(*args, **kw, flush=True)
