"""Leveled logging (reference common/logging.h:26-105 equivalent).

``BPS_LOG_LEVEL`` ∈ {TRACE, DEBUG, INFO, WARNING, ERROR}; ``BPS_LOG_HIDE_TIME``
drops timestamps (reference BYTEPS_LOG_HIDE_TIME).
"""

import logging
import os
import sys

TRACE = 5
logging.addLevelName(TRACE, "TRACE")

_logger = None


def get_logger() -> logging.Logger:
    global _logger
    if _logger is not None:
        return _logger
    logger = logging.getLogger("byteps_amd")
    level_name = os.environ.get(
        "BPS_LOG_LEVEL", os.environ.get("BYTEPS_LOG_LEVEL", "INFO")).upper()
    level = TRACE if level_name == "TRACE" else getattr(logging, level_name, logging.INFO)
    logger.setLevel(level)
    if not logger.handlers:
        h = logging.StreamHandler(sys.stderr)
        hide_time = os.environ.get(
            "BPS_LOG_HIDE_TIME", os.environ.get("BYTEPS_LOG_HIDE_TIME", "0")) == "1"
        fmt = "[%(levelname)s bps] %(message)s" if hide_time else \
              "[%(asctime)s %(levelname)s bps] %(message)s"
        h.setFormatter(logging.Formatter(fmt))
        logger.addHandler(h)
    logger.propagate = False
    _logger = logger
    return logger
