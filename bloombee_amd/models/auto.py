"""AutoDistributed* dispatch (parity: reference utils/auto_config.py:82-99)."""
from __future__ import annotations

from bloombee_amd.models.base import ModelConfig, get_family, resolve_config


class AutoDistributedConfig:
    @staticmethod
    def from_pretrained(name_or_path: str, **kw) -> ModelConfig:
        cfg = resolve_config(name_or_path)
        for k, v in kw.items():
            if hasattr(cfg, k):
                setattr(cfg, k, v)
            else:
                cfg.extras[k] = v
        return cfg


class _AutoModelBase:
    _attr = "model_cls"

    @classmethod
    def from_pretrained(cls, name_or_path: str, **kw):
        cfg = resolve_config(name_or_path)
        family = get_family(cfg.model_type)
        model_cls = getattr(family, cls._attr)
        if model_cls is None:
            raise NotImplementedError(
                f"family {cfg.model_type!r} has no {cls._attr} registered yet")
        return model_cls.from_pretrained(name_or_path, config=cfg, **kw)


class AutoDistributedModel(_AutoModelBase):
    _attr = "model_cls"


class AutoDistributedModelForCausalLM(_AutoModelBase):
    _attr = "causal_lm_cls"


class AutoDistributedSpeculativeModel(_AutoModelBase):
    _attr = "speculative_cls"


class AutoDistributedModelForSequenceClassification(_AutoModelBase):
    _attr = "seq_cls_cls"


def get_block_class(model_type: str):
    return get_family(model_type).block_cls
