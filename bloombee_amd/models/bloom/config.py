"""BLOOM family config + presets (parity: reference models/bloom/config.py).

BLOOM is the reference's canonical plumbing family — BASELINE.json config 1
is bloom-560m on two local CPU workers over loopback DHT.
"""
from __future__ import annotations

from dataclasses import dataclass

from bloombee_amd.models.base import ModelConfig

BLOOM_PRESETS = {
    "bigscience/bloom-560m": dict(
        hidden_size=1024, num_hidden_layers=24, num_attention_heads=16,
        intermediate_size=4096, vocab_size=250880, tie_word_embeddings=True,
        layer_norm_epsilon=1e-5, max_position_embeddings=2048,
    ),
    "bloom-560m": dict(
        hidden_size=1024, num_hidden_layers=24, num_attention_heads=16,
        intermediate_size=4096, vocab_size=250880, tie_word_embeddings=True,
        layer_norm_epsilon=1e-5, max_position_embeddings=2048,
    ),
    "bloom-tiny": dict(
        hidden_size=128, num_hidden_layers=4, num_attention_heads=4,
        intermediate_size=512, vocab_size=1024, tie_word_embeddings=True,
        max_position_embeddings=2048,
    ),
}


@dataclass
class BloomConfig(ModelConfig):
    model_type: str = "bloom"

    @classmethod
    def from_dict(cls, d: dict) -> "BloomConfig":
        d = dict(d)
        # HF bloom config.json key aliases
        if "n_layer" in d:
            d.setdefault("num_hidden_layers", d.pop("n_layer"))
        if "n_head" in d:
            d.setdefault("num_attention_heads", d.pop("n_head"))
        if "n_embed" in d:
            d.setdefault("hidden_size", d.pop("n_embed"))
        d.setdefault("intermediate_size", 4 * d.get("hidden_size", 1024))
        return super().from_dict(d)
