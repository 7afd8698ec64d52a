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
    # qwen3 qk-norm (blocks without the attribute skip these)
    "q_norm_w": "self_attn.q_norm.weight",
    "k_norm_w": "self_attn.k_norm.weight",
}


def _save_tensor(path: Path, t: torch.Tensor) -> None:
    t = t.detach().contiguous().cpu()
    if t.dtype == torch.bfloat16:
        np.save(str(path) + ".bf16.npy", t.view(torch.uint16).numpy())
    else:
        # dtype-preserving (npy is self-describing): fp32 checkpoints stay
        # lossless; fp16 stays half-size like the reference layout
        np.save(str(path) + ".npy", t.numpy())


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


# ---------------------------------------------------------------------------
# Real-checkpoint conversion: HF model directory -> per-block npy layout
# ---------------------------------------------------------------------------
def _iter_hf_state_dict(hf_dir: str):
    """Yield (name, tensor) from a local HF checkpoint without instantiating
    the model: safetensors shards preferred (streamed one tensor at a time),
    pytorch_model.bin fallback (whole-shard torch.load).

    Reference semantics: server/from_pretrained.py:59-130 loads per-block
    state dicts from downloaded HF shards; flexgen_utils/llama_config.py:
    150-241 converts them to the per-tensor npy layout. This environment has
    no hub egress, so only local directories are supported."""
    import glob
    import json as _json

    d = Path(hf_dir)
    st_index = d / "model.safetensors.index.json"
    st_single = d / "model.safetensors"
    if st_index.exists() or st_single.exists():
        from safetensors import safe_open

        if st_index.exists():
            with open(st_index) as f:
                shards = sorted(set(_json.load(f)["weight_map"].values()))
        else:
            shards = [st_single.name]
        for shard in shards:
            with safe_open(str(d / shard), framework="pt", device="cpu") as f:
                for name in f.keys():
                    yield name, f.get_tensor(name)
        return
    bin_index = d / "pytorch_model.bin.index.json"
    if bin_index.exists():
        with open(bin_index) as f:
            shards = sorted(set(_json.load(f)["weight_map"].values()))
    elif (d / "pytorch_model.bin").exists():
        shards = ["pytorch_model.bin"]
    else:
        raise FileNotFoundError(
            f"no safetensors/bin checkpoint under {hf_dir}")
    for shard in shards:
        sd = torch.load(d / shard, map_location="cpu", weights_only=True)
        yield from sd.items()


def convert_hf_checkpoint(hf_dir: str, out_dir: Optional[str] = None,
                          force: bool = False) -> str:
    """Convert a local HF checkpoint (llama/qwen3-pattern naming) into the
    per-block npy layout this framework serves from. Returns the output dir
    (``{hf_dir}-np`` by default, mirroring the reference's ``{model}-np``).

    Fusions performed at conversion time (the serve-time layout):
      * q/k/v projections -> one ``self_attn.qkv_proj.weight``
      * gate/up           -> one ``mlp.gate_up_proj.weight``
    Client tensors (embed/norm/lm_head) are written alongside; config.json
    is copied so ``resolve_config(out_dir)`` works; a sentinel marks
    completion (partial conversions re-run).
    """
    import json as _json
    import shutil

    d = Path(hf_dir)
    out = Path(out_dir) if out_dir else Path(str(d).rstrip("/") + "-np")
    if is_converted(str(out)) and not force:
        return str(out)
    out.mkdir(parents=True, exist_ok=True)

    pending: Dict[str, Dict[str, torch.Tensor]] = {}

    def _fuse(idx: int, group: str, key: str, t: torch.Tensor,
              want: tuple, stem: str) -> None:
        slot = pending.setdefault(f"{idx}.{group}", {})
        slot[key] = t
        if all(k in slot for k in want):
            fused = torch.cat([slot[k] for k in want], dim=0)
            _save_tensor(out / f"layers.{idx}.{stem}", fused)
            del pending[f"{idx}.{group}"]

    n_tensors = 0
    tied_lm_head = True
    for name, t in _iter_hf_state_dict(str(d)):
        n_tensors += 1
        name = name.removeprefix("model.")
        if name == "embed_tokens.weight":
            _save_tensor(out / "embed_tokens.weight", t)
        elif name == "norm.weight":
            _save_tensor(out / "norm.weight", t)
        elif name == "lm_head.weight":
            tied_lm_head = False
            _save_tensor(out / "lm_head.weight", t)
        elif name.startswith("layers."):
            _, i, rest = name.split(".", 2)
            idx = int(i)
            if rest in ("self_attn.q_proj.weight", "self_attn.k_proj.weight",
                        "self_attn.v_proj.weight"):
                _fuse(idx, "qkv", rest.split(".")[1][0], t, ("q", "k", "v"),
                      "self_attn.qkv_proj.weight")
            elif rest in ("mlp.gate_proj.weight", "mlp.up_proj.weight"):
                _fuse(idx, "gu", "gate" if "gate" in rest else "up", t,
                      ("gate", "up"), "mlp.gate_up_proj.weight")
            else:
                _save_tensor(out / f"layers.{idx}.{rest}", t)
        else:
            # rotary inv_freq buffers etc. — not weights
            continue
    if pending:
        raise ValueError(f"incomplete fused groups after conversion: "
                         f"{sorted(pending)}")
    if n_tensors == 0:
        raise ValueError(f"no tensors found in {hf_dir}")
    # config + tokenizer files travel with the converted layout
    for fname in ("config.json", "tokenizer.json", "tokenizer_config.json",
                  "special_tokens_map.json", "tokenizer.model"):
        src = d / fname
        if src.exists():
            shutil.copy(src, out / fname)
    cfg_path = out / "config.json"
    if cfg_path.exists() and tied_lm_head:
        with open(cfg_path) as f:
            cfg = _json.load(f)
        cfg["tie_word_embeddings"] = True
        with open(cfg_path, "w") as f:
            _json.dump(cfg, f)
    (out / SENTINEL).touch()
    return str(out)
