"""Singleton stderr logger for the framework.

Parity: reference backend/utils/logging.py:10-35 (singleton "dts" logger,
fixed format, level from env).
"""

from __future__ import annotations

import logging
import os
import sys


def _build_logger() -> logging.Logger:
    log = logging.getLogger("dts_amd")
    if log.handlers:
        return log
    level_name = os.environ.get("DTS_LOG_LEVEL", "INFO").upper()
    level = getattr(logging, level_name, logging.INFO)
    handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(
        logging.Formatter("%(asctime)s [%(levelname)s] %(name)s: %(message)s", "%H:%M:%S")
    )
    log.addHandler(handler)
    log.setLevel(level)
    log.propagate = False
    return log


logger = _build_logger()


def log_phase(phase: str, message: str, indent: int = 0) -> None:
    """Structured phase log line, `[DTS:PHASE] msg` (ref core/dts/utils.py:14-30)."""
    logger.info("[DTS:%s] %s%s", phase, "  " * indent, message)
