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
        # per-module overrides: BBAMD_LOG="client.session=debug,server=warning"
        # (parity: reference debug_config per-group toggles, generalized to
        # any logger subtree)
        for spec in filter(None, os.environ.get("BBAMD_LOG", "").split(",")):
            mod, _, lvl = spec.partition("=")
            if mod and lvl:
                logging.getLogger(f"bloombee_amd.{mod.strip()}") \
                    .setLevel(lvl.strip().upper())
        _configured = True
    return logging.getLogger(name if name.startswith("bloombee_amd") else f"bloombee_amd.{name}")


# ---------------------------------------------------------------------------
# Per-channel debug switches (parity: reference utils/debug_config.py —
# BLOOMBEE_DEBUG master + per-group toggles). BBAMD_DEBUG=1 enables all;
# BBAMD_DEBUG_KV / _MICROBATCH / _COMPRESSION / _INFERENCE narrow it.
# ---------------------------------------------------------------------------
import os as _os

_MASTER = _os.environ.get("BBAMD_DEBUG", "") not in ("", "0", "false")
_CHANNELS = {
    ch: _os.environ.get(f"BBAMD_DEBUG_{ch.upper()}", "") not in ("", "0", "false")
    for ch in ("kv", "microbatch", "compression", "inference")
}


def is_log_channel_enabled(channel: str) -> bool:
    return _MASTER or _CHANNELS.get(channel, False)


def debug_log(channel: str, logger, msg: str, *args) -> None:
    if is_log_channel_enabled(channel):
        logger.info(f"[{channel}] {msg}", *args)
