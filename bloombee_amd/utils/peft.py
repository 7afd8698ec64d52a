"""LoRA adapters for server blocks (parity: reference utils/peft.py:133-251 —
`using_adapter` contextvar, LoraLinear swap-in, per-adapter weight sets; the
HF-hub download path is replaced by local .npy adapter dirs, matching the
offline per-block checkpoint layout)."""
from __future__ import annotations

import contextlib
import contextvars
import math
import os
from pathlib import Path
from typing import Dict, List, Optional

import numpy as np
import torch

from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)

_active_adapter: contextvars.ContextVar = contextvars.ContextVar(
    "bbamd_active_adapter", default=None)


@contextlib.contextmanager
def using_adapter(name: Optional[str]):
    """Activate adapter `name` for ops inside the context (ref
    AdapterContextMixin.using_adapter)."""
    tok = _active_adapter.set(name)
    try:
        yield
    finally:
        _active_adapter.reset(tok)


def active_adapter() -> Optional[str]:
    return _active_adapter.get()


class LoraSet:
    """One adapter's (A, B) pair for one base parameter."""

    def __init__(self, a: torch.Tensor, b: torch.Tensor, alpha: float = 1.0):
        self.a, self.b = a, b          # a: (r, in), b: (out, r)
        self.scale = alpha / a.shape[0]

    def delta(self, x: torch.Tensor) -> torch.Tensor:
        return (x @ self.a.t().to(x.dtype)) @ self.b.t().to(x.dtype) * self.scale


class LoraAdapterMixin:
    """Adds per-parameter LoRA deltas to a block. Attach with
    `add_adapter_to_block`; the block's ops.linear calls route through
    `lora_linear` when an adapter is active."""

    def init_lora(self):
        if not hasattr(self, "_lora"):
            self._lora: Dict[str, Dict[str, LoraSet]] = {}

    def add_adapter(self, name: str, sets: Dict[str, LoraSet]):
        self.init_lora()
        self._lora[name] = sets

    def lora_delta(self, param_name: str, x: torch.Tensor) -> Optional[torch.Tensor]:
        name = active_adapter()
        if name is None or not getattr(self, "_lora", None):
            return None
        sets = self._lora.get(name)
        if sets is None or param_name not in sets:
            return None
        return sets[param_name].delta(x)


def create_lora_adapter(block: torch.nn.Module, rank: int = 8,
                        alpha: float = 16.0, targets: Optional[List[str]] = None,
                        seed: int = 0) -> Dict[str, LoraSet]:
    """Fresh random-A/zero-B adapter for a block's linear weights (ref
    create_lora_adapter :186-250)."""
    targets = targets or ["qkv_w", "o_w"]
    gen = torch.Generator().manual_seed(seed)
    sets = {}
    for t in targets:
        w = getattr(block, t, None)
        if w is None:
            continue
        out_f, in_f = w.shape
        a = torch.randn(rank, in_f, generator=gen) / math.sqrt(in_f)
        b = torch.zeros(out_f, rank)
        sets[t] = LoraSet(a.to(w.dtype), b.to(w.dtype), alpha)
    return sets


def save_adapter(sets: Dict[str, LoraSet], path: str) -> None:
    d = Path(path)
    d.mkdir(parents=True, exist_ok=True)
    for pname, s in sets.items():
        np.save(d / f"{pname}.lora_a.npy", s.a.float().numpy())
        np.save(d / f"{pname}.lora_b.npy", s.b.float().numpy())


def load_adapter(path: str, dtype=torch.bfloat16, alpha: float = 16.0,
                 ) -> Dict[str, LoraSet]:
    d = Path(path)
    sets = {}
    for fa in d.glob("*.lora_a.npy"):
        pname = fa.name[: -len(".lora_a.npy")]
        a = torch.from_numpy(np.load(fa)).to(dtype)
        b = torch.from_numpy(np.load(d / f"{pname}.lora_b.npy")).to(dtype)
        sets[pname] = LoraSet(a, b, alpha)
    return sets


def estimate_adapter_memory(block: torch.nn.Module, rank: int,
                            targets: Optional[List[str]] = None) -> int:
    """Bytes one adapter adds per block (ref peft.py:251)."""
    targets = targets or ["qkv_w", "o_w"]
    total = 0
    for t in targets:
        w = getattr(block, t, None)
        if w is not None:
            out_f, in_f = w.shape
            total += (rank * in_f + out_f * rank) * w.element_size()
    return total


def add_adapter_to_block(block: torch.nn.Module, name: str,
                         sets: Dict[str, LoraSet]) -> None:
    """Attach + monkey-wrap the block's linear calls. Blocks built on
    ops.linear pick deltas up through `lora_linear`."""
    if not isinstance(block, LoraAdapterMixin):
        block.__class__ = type(block.__class__.__name__ + "WithLora",
                               (LoraAdapterMixin, block.__class__), {})
    # adapters are trained against the unfolded parameterization — undo the
    # fused-norm weight fold (llama/block.py fold_norm_weights) if applied
    if getattr(block, "_norm_folded", False):
        block.unfold_norm_weights()
    block.init_lora()
    block.add_adapter(name, sets)
