"""User-facing logging (reference: bodo/user_logging.py — global logger +
integer verbose level; compiler/optimizer diagnostics route through it)."""

from __future__ import annotations

import logging
import sys
from typing import Optional

_LOGGER: Optional[logging.Logger] = None
_VERBOSE_LEVEL = 0


def get_current_bodo_verbose_level() -> int:
    return _VERBOSE_LEVEL


def set_verbose_level(level: int):
    global _VERBOSE_LEVEL
    assert level >= 0
    _VERBOSE_LEVEL = level


def get_verbose_logger() -> logging.Logger:
    global _LOGGER
    if _LOGGER is None:
        _LOGGER = logging.getLogger("bodo_amd")
        if not _LOGGER.handlers:
            h = logging.StreamHandler(sys.stdout)
            h.setFormatter(logging.Formatter("%(message)s"))
            _LOGGER.addHandler(h)
        _LOGGER.setLevel(logging.INFO)
    return _LOGGER


def set_bodo_verbose_logger(logger: logging.Logger):
    global _LOGGER
    _LOGGER = logger


def log_message(header: str, msg: str, *args, level: int = 1):
    """Log when verbose level >= level, rank 0 only (reference semantics)."""
    from .parallel import comm

    if _VERBOSE_LEVEL >= level and comm.get_rank() == 0:
        logger = get_verbose_logger()
        logger.info("%s:\n%s", header, msg % args if args else msg)
