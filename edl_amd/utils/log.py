"""Logging helpers (capability parity: reference utils/log_utils.py:16-32)."""
import logging
import os
import sys

_FMT = "%(asctime)s %(levelname)s [%(name)s] %(message)s"


def get_logger(name="edl", level=None):
    """Return a configured logger. Level from arg, $EDL_LOG_LEVEL, or INFO."""
    if level is None:
        level = os.environ.get("EDL_LOG_LEVEL", "INFO")
    if isinstance(level, str):
        level = getattr(logging, level.upper(), logging.INFO)
    logger = logging.getLogger(name)
    logger.setLevel(level)
    if not logger.handlers:
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(logging.Formatter(_FMT))
        logger.addHandler(h)
        logger.propagate = False
    return logger
