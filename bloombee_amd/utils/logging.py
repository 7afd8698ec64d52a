"""Logging helpers (reference: src/bloombee/utils/logging.py)."""
from __future__ import annotations

import logging
import os
import sys

_FMT = "%(asctime)s.%(msecs)03d [%(levelname).1s] %(name)s: %(message)s"
_DATEFMT = "%H:%M:%S"
_configured = False


def get_logger(name: str) -> logging.Logger:
    global _configured
    if not _configured:
        level = os.environ.get("BBAMD_LOGLEVEL", "INFO").upper()
        handler = logging.StreamHandler(sys.stderr)
        handler.setFormatter(logging.Formatter(_FMT, _DATEFMT))
        root = logging.getLogger("bloombee_amd")
        root.addHandler(handler)
        root.setLevel(level)
        root.propagate = False
        _configured = True
    return logging.getLogger(name if name.startswith("bloombee_amd") else f"bloombee_amd.{name}")
