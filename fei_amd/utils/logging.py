"""Logging for fei_amd.

Behavioral parity with the reference logging layer (fei/utils/logging.py:12-118):
env-configured level/file (``FEI_LOG_LEVEL`` / ``FEI_LOG_FILE``), cached
per-name loggers, and a rotating file handler (10 MB x 5).
"""

from __future__ import annotations

import logging
import logging.handlers
import os
import sys
import threading
from typing import Dict, Optional

_LOGGERS: Dict[str, logging.Logger] = {}
_LOCK = threading.Lock()
_CONFIGURED = False

_FORMAT = "%(asctime)s %(levelname)s %(name)s: %(message)s"


def _level_from_env(default: str = "WARNING") -> int:
    name = os.environ.get("FEI_LOG_LEVEL", default).upper()
    return getattr(logging, name, logging.WARNING)


def setup_logging(
    level: Optional[int] = None,
    log_file: Optional[str] = None,
    stream=None,
) -> None:
    """Configure the root ``fei_amd`` logger once.

    Args:
        level: numeric level; defaults to ``FEI_LOG_LEVEL`` env (WARNING).
        log_file: path for a rotating file handler; defaults to
            ``FEI_LOG_FILE`` env (no file logging if unset).
        stream: stream for the console handler (stderr by default).
    """
    global _CONFIGURED
    with _LOCK:
        root = logging.getLogger("fei_amd")
        if level is None:
            level = _level_from_env()
        root.setLevel(level)

        if _CONFIGURED:
            return

        handler = logging.StreamHandler(stream or sys.stderr)
        handler.setFormatter(logging.Formatter(_FORMAT))
        root.addHandler(handler)

        log_file = log_file or os.environ.get("FEI_LOG_FILE")
        if log_file:
            try:
                os.makedirs(os.path.dirname(os.path.abspath(log_file)), exist_ok=True)
                fh = logging.handlers.RotatingFileHandler(
                    log_file, maxBytes=10 * 1024 * 1024, backupCount=5
                )
                fh.setFormatter(logging.Formatter(_FORMAT))
                root.addHandler(fh)
            except OSError:
                root.warning("could not open log file %s", log_file)

        root.propagate = False
        _CONFIGURED = True


def get_logger(name: str) -> logging.Logger:
    """Return (and cache) a child logger under the ``fei_amd`` namespace."""
    with _LOCK:
        if name in _LOGGERS:
            return _LOGGERS[name]
    setup_logging()
    if not name.startswith("fei_amd"):
        name = "fei_amd." + name
    logger = logging.getLogger(name)
    with _LOCK:
        _LOGGERS[name] = logger
    return logger
