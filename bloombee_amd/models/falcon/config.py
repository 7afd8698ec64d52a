"""Falcon family config + presets (parity: reference models/falcon/)."""
from __future__ import annotations

from dataclasses import dataclass

from bloombee_amd.models.base import ModelConfig

FALCON_PRESETS = {
    "tiiuae/falcon-7b": dict(
        hidden_size=4544, num_hidden_layers=32, num_attention_heads=71,
        num_key_value_heads=1, head_dim=64, intermediate_size=4 * 4544,
        vocab_size=65024, tie_word_embeddings=True, layer_norm_epsilon=1e-5,
        rope_theta=10000.0, max_position_embeddings=2048,
    ),
    "falcon-7b": dict(
        hidden_size=4544, num_hidden_layers=32, num_attention_heads=71,
        num_key_value_heads=1, head_dim=64, intermediate_size=4 * 4544,
        vocab_size=65024, tie_word_embeddings=True, layer_norm_epsilon=1e-5,
        rope_theta=10000.0, max_position_embeddings=2048,
    ),
    "falcon-tiny": dict(
        hidden_size=256, num_hidden_layers=4, num_attention_heads=8,
        num_key_value_heads=1, head_dim=32, intermediate_size=1024,
        vocab_size=1024, tie_word_embeddings=True, max_position_embeddings=2048,
    ),
}


@dataclass
class FalconConfig(ModelConfig):
    model_type: str = "falcon"

    @classmethod
    def from_dict(cls, d: dict) -> "FalconConfig":
        d = dict(d)
        if "n_layer" in d:
            d.setdefault("num_hidden_layers", d.pop("n_layer"))
        if "n_head" in d:
            d.setdefault("num_attention_heads", d.pop("n_head"))
        if d.pop("multi_query", False):
            d.setdefault("num_key_value_heads", 1)
        d.setdefault("intermediate_size", 4 * d.get("hidden_size", 4544))
        return super().from_dict(d)
