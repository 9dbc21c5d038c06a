"""Leveled logger (reference include/logging/logger.hpp:16 spdlog analog)."""

from __future__ import annotations

import logging
import os
import sys

_FORMAT = "[%(asctime)s] [%(name)s] [%(levelname)s] %(message)s"
_configured = False


def _configure():
    global _configured
    if _configured:
        return
    level = os.environ.get("TNN_LOG_LEVEL", "INFO").upper()
    handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(logging.Formatter(_FORMAT, datefmt="%H:%M:%S"))
    root = logging.getLogger("tnn")
    root.setLevel(getattr(logging, level, logging.INFO))
    root.addHandler(handler)
    root.propagate = False
    _configured = True


def get_logger(name: str = "") -> logging.Logger:
    _configure()
    return logging.getLogger(f"tnn.{name}" if name else "tnn")
