"""Per-block checkpoint IO in the reference's `.npy` weight layout.

The north star (BASELINE.json) requires preserving the reference's per-block
checkpoint layout: one fp16/bf16 `.npy` file per tensor at
``{path}/{model}-np/layers.{i}.{param}`` plus embeddings/norm/lm_head files,
with a ``.bloombee_np_converted`` sentinel (reference
flexgen_utils/llama_config.py:150-241 `download_llama_weights`).

bf16 has no numpy dtype, so bf16 tensors are stored as uint16 views with a
``.bf16.npy`` suffix — lossless, self-describing, still one file per tensor.
"""
from __future__ import annotations

import os
from pathlib import Path
from typing import Dict, Optional

import numpy as np
import torch

SENTINEL = ".bloombee_np_converted"

# our flat parameter names -> reference file-name stems (per block)
_BLOCK_PARAM_MAP = {
    "input_norm_w": "input_layernorm.weight",
    "qkv_w": "self_attn.qkv_proj.weight",        # fused — split files also supported
    "o_w": "self_attn.o_proj.weight",
    "post_norm_w": "post_attention_layernorm.weight",
    "gate_up_w": "mlp.gate_up_proj.weight",      # fused
    "down_w": "mlp.down_proj.weight",
}


def _save_tensor(path: Path, t: torch.Tensor) -> None:
    t = t.detach().contiguous().cpu()
    if t.dtype == torch.bfloat16:
        np.save(str(path) + ".bf16.npy", t.view(torch.uint16).numpy())
    else:
        np.save(str(path) + ".npy", t.to(torch.float16).numpy())


def _load_tensor(path: Path) -> Optional[torch.Tensor]:
    bf = Path(str(path) + ".bf16.npy")
    if bf.exists():
        return torch.from_numpy(np.load(bf)).view(torch.bfloat16)
    fp = Path(str(path) + ".npy")
    if fp.exists():
        return torch.from_numpy(np.load(fp))
    return None


def save_block_weights(block: torch.nn.Module, ckpt_dir: str, index: int) -> None:
    d = Path(ckpt_dir)
    d.mkdir(parents=True, exist_ok=True)
    for pname, stem in _BLOCK_PARAM_MAP.items():
        if hasattr(block, pname):
            _save_tensor(d / f"layers.{index}.{stem}", getattr(block, pname))
    (d / SENTINEL).touch()


def load_block_weights(block: torch.nn.Module, ckpt_dir: str, index: int) -> int:
    """Load one block's tensors from the npy layout; returns #tensors loaded.
    Supports the fused qkv/gate_up layout directly and falls back to
    concatenating the reference's split q/k/v and gate/up files."""
    d = Path(ckpt_dir)
    loaded = 0
    for pname, stem in _BLOCK_PARAM_MAP.items():
        if not hasattr(block, pname):
            continue
        t = _load_tensor(d / f"layers.{index}.{stem}")
        if t is None and pname == "qkv_w":
            parts = [_load_tensor(d / f"layers.{index}.self_attn.{w}_proj.weight")
                     for w in ("q", "k", "v")]
            if all(p is not None for p in parts):
                t = torch.cat(parts, dim=0)
        if t is None and pname == "gate_up_w":
            parts = [_load_tensor(d / f"layers.{index}.mlp.{w}_proj.weight")
                     for w in ("gate", "up")]
            if all(p is not None for p in parts):
                t = torch.cat(parts, dim=0)
        if t is not None:
            with torch.no_grad():
                getattr(block, pname).copy_(t.to(getattr(block, pname).dtype))
            loaded += 1
    return loaded


def save_client_weights(ckpt_dir: str, embed: torch.Tensor, final_norm: torch.Tensor,
                        lm_head: Optional[torch.Tensor]) -> None:
    d = Path(ckpt_dir)
    d.mkdir(parents=True, exist_ok=True)
    _save_tensor(d / "embed_tokens.weight", embed)
    _save_tensor(d / "norm.weight", final_norm)
    if lm_head is not None:
        _save_tensor(d / "lm_head.weight", lm_head)


def load_client_weights(ckpt_dir: str) -> Dict[str, Optional[torch.Tensor]]:
    d = Path(ckpt_dir)
    return {
        "embed": _load_tensor(d / "embed_tokens.weight"),
        "final_norm": _load_tensor(d / "norm.weight"),
        "lm_head": _load_tensor(d / "lm_head.weight"),
    }


def is_converted(ckpt_dir: str) -> bool:
    return (Path(ckpt_dir) / SENTINEL).exists()
