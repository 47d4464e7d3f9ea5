"""Logging setup (reference: pkg/log/log.go — zap + lumberjack rotation).

Python analog: stdlib logging with an optional rotating file handler. The
global ``logger`` is swappable like the reference's package-level Logger.
"""

from __future__ import annotations

import logging
import logging.handlers
import sys
from typing import Optional

_FMT = "%(asctime)s\t%(levelname)s\t%(name)s\t%(message)s"

logger = logging.getLogger("gpud")


def setup(
    level: str = "info",
    log_file: Optional[str] = None,
    max_bytes: int = 64 * 1024 * 1024,
    backup_count: int = 5,
) -> logging.Logger:
    """Configure the global gpud logger. Safe to call more than once."""
    lvl = getattr(logging, level.upper(), logging.INFO)
    logger.setLevel(lvl)
    logger.handlers.clear()
    fmt = logging.Formatter(_FMT)
    sh = logging.StreamHandler(sys.stderr)
    sh.setFormatter(fmt)
    logger.addHandler(sh)
    if log_file:
        fh = logging.handlers.RotatingFileHandler(
            log_file, maxBytes=max_bytes, backupCount=backup_count
        )
        fh.setFormatter(fmt)
        logger.addHandler(fh)
    logger.propagate = False
    return logger


def audit_logger(log_file: Optional[str] = None) -> logging.Logger:
    """Separate audit logger (reference: pkg/log audit variant)."""
    lg = logging.getLogger("gpud.audit")
    if not lg.handlers:
        fmt = logging.Formatter(_FMT)
        h: logging.Handler
        if log_file:
            h = logging.handlers.RotatingFileHandler(
                log_file, maxBytes=16 * 1024 * 1024, backupCount=3
            )
        else:
            h = logging.StreamHandler(sys.stderr)
        h.setFormatter(fmt)
        lg.addHandler(h)
        lg.propagate = False
    return lg
