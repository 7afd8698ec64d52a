"""Hidden-state capture for compression research (parity: reference
utils/real_activation_dumper.py:1-345 `capture_activation` — opt-in hook at
the backend that dumps real per-step activations to disk for offline wire-
codec studies). Enabled via BBAMD_DUMP_ACTIVATIONS=/path."""
from __future__ import annotations

import os
import time
from pathlib import Path
from typing import Optional

import torch

from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)

_DIR = os.environ.get("BBAMD_DUMP_ACTIVATIONS")
_counter = 0


def enabled() -> bool:
    return _DIR is not None


def capture_activation(tag: str, tensor: torch.Tensor,
                       step: Optional[int] = None) -> Optional[str]:
    """Dump one activation tensor (cpu, fp16) under the configured dir.
    Returns the path, or None when disabled."""
    global _counter
    if _DIR is None:
        return None
    d = Path(_DIR)
    d.mkdir(parents=True, exist_ok=True)
    _counter += 1
    name = f"{tag}_s{step if step is not None else _counter}_{_counter}.pt"
    path = d / name
    torch.save(tensor.detach().to(torch.float16).cpu(), path)
    if _counter <= 3:
        logger.info("captured activation %s %s", name, tuple(tensor.shape))
    return str(path)
