"""Model family base classes and registry.

Mirrors the reference registration pattern: each family
provides a config (HF-config-compatible field names, loadable from a local
config.json), a server-side block class built on the gfx950 op library, and
a client-side distributed model class. Families register via
``register_model_family`` and are dispatched by ``model_type``
(ref utils/auto_config.py:25-99).
"""
from __future__ import annotations

import dataclasses
import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, Optional, Type

import torch


@dataclass
class ModelConfig:
    """Architecture description. Field names follow HF conventions so a local
    checkpoint's config.json loads directly (no network in this environment —
    named presets provide random-init shapes)."""

    model_type: str = "llama"
    hidden_size: int = 4096
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: Optional[int] = None
    head_dim: Optional[int] = None
    intermediate_size: int = 14336
    vocab_size: int = 128256
    max_position_embeddings: int = 8192
    rms_norm_eps: float = 1e-5
    layer_norm_epsilon: float = 1e-5
    rope_theta: float = 500000.0
    rope_scaling: Optional[dict] = None
    tie_word_embeddings: bool = False
    torch_dtype: str = "bfloat16"
    # family-specific extras (sliding windows, MoE, alibi...) ride here
    extras: Dict[str, Any] = field(default_factory=dict)

    def __post_init__(self):
        if self.num_key_value_heads is None:
            self.num_key_value_heads = self.num_attention_heads
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_attention_heads

    @property
    def dtype(self) -> torch.dtype:
        # BBAMD_TORCH_DTYPE (run_server --torch-dtype) overrides the preset
        import os
        return getattr(torch, os.environ.get("BBAMD_TORCH_DTYPE",
                                             self.torch_dtype))

    @classmethod
    def from_dict(cls, d: dict) -> "ModelConfig":
        d = dict(d)
        # transformers >=5 config spellings: "dtype" replaced "torch_dtype",
        # rope params moved under "rope_parameters"
        if "dtype" in d and "torch_dtype" not in d:
            d["torch_dtype"] = d.pop("dtype")
        rp = d.get("rope_parameters")
        if isinstance(rp, dict):
            d.setdefault("rope_theta", rp.get("rope_theta", 10000.0))
            if rp.get("rope_type", "default") not in ("default", None):
                d.setdefault("rope_scaling", rp)
        known = {f.name for f in dataclasses.fields(cls)}
        kw = {k: v for k, v in d.items() if k in known}
        extras = {k: v for k, v in d.items() if k not in known}
        cfg = cls(**kw)
        cfg.extras.update(extras)
        return cfg

    def to_dict(self) -> dict:
        d = dataclasses.asdict(self)
        d.update(d.pop("extras"))
        return d

    @classmethod
    def from_json(cls, path: str) -> "ModelConfig":
        with open(path) as f:
            return cls.from_dict(json.load(f))


@dataclass
class FamilyEntry:
    config_cls: Type[ModelConfig]
    block_cls: type
    model_cls: Optional[type] = None       # client DistributedModel
    causal_lm_cls: Optional[type] = None   # client DistributedModelForCausalLM
    speculative_cls: Optional[type] = None # client ForSpeculativeGeneration
    seq_cls_cls: Optional[type] = None     # client ForSequenceClassification
    presets: Dict[str, dict] = field(default_factory=dict)


_FAMILIES: Dict[str, FamilyEntry] = {}
_PRESETS: Dict[str, tuple] = {}  # preset name -> (model_type, dict)


def register_model_family(model_type: str, entry: FamilyEntry) -> None:
    _FAMILIES[model_type] = entry
    for name, d in entry.presets.items():
        _PRESETS[name.lower()] = (model_type, d)


def get_family(model_type: str) -> FamilyEntry:
    if model_type not in _FAMILIES:
        raise KeyError(
            f"unknown model family {model_type!r}; registered: {sorted(_FAMILIES)}"
        )
    return _FAMILIES[model_type]


def resolve_config(name_or_path: str) -> ModelConfig:
    """Resolve a model name or local path to a ModelConfig.

    Order: local directory with config.json -> registered preset name.
    (The reference downloads from HF hub; this environment has no egress, so
    presets carry the published architecture shapes for random-init runs.)
    """
    cfg_path = os.path.join(name_or_path, "config.json")
    if os.path.isfile(cfg_path):
        with open(cfg_path) as f:
            d = json.load(f)
        family = get_family(d.get("model_type", "llama"))
        return family.config_cls.from_dict(d)
    key = name_or_path.lower()
    if key in _PRESETS:
        model_type, d = _PRESETS[key]
        family = get_family(model_type)
        return family.config_cls.from_dict(dict(d, model_type=model_type))
    raise ValueError(
        f"cannot resolve model {name_or_path!r}: not a local checkpoint dir and "
        f"not a known preset ({sorted(_PRESETS)})"
    )
