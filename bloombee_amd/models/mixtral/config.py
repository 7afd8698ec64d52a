"""Mixtral family config + presets (parity: reference models/mixtral/)."""
from __future__ import annotations

from dataclasses import dataclass

from bloombee_amd.models.base import ModelConfig

MIXTRAL_PRESETS = {
    "mistralai/mixtral-8x7b-v0.1": dict(
        hidden_size=4096, num_hidden_layers=32, num_attention_heads=32,
        num_key_value_heads=8, intermediate_size=14336, vocab_size=32000,
        rms_norm_eps=1e-5, rope_theta=1000000.0, max_position_embeddings=32768,
        num_local_experts=8, num_experts_per_tok=2,
    ),
    "mixtral-8x7b": dict(
        hidden_size=4096, num_hidden_layers=32, num_attention_heads=32,
        num_key_value_heads=8, intermediate_size=14336, vocab_size=32000,
        rms_norm_eps=1e-5, rope_theta=1000000.0, max_position_embeddings=32768,
        num_local_experts=8, num_experts_per_tok=2,
    ),
    # 4-layer shard of the 8x7b shape for single-GPU config evidence
    "mixtral-8x7b-4l": dict(
        hidden_size=4096, num_hidden_layers=4, num_attention_heads=32,
        num_key_value_heads=8, intermediate_size=14336, vocab_size=32000,
        rms_norm_eps=1e-5, rope_theta=1000000.0, max_position_embeddings=32768,
        num_local_experts=8, num_experts_per_tok=2,
    ),
    "mixtral-tiny": dict(
        hidden_size=256, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, intermediate_size=512, vocab_size=1024,
        max_position_embeddings=2048, num_local_experts=4,
        num_experts_per_tok=2,
    ),
}


@dataclass
class MixtralConfig(ModelConfig):
    model_type: str = "mixtral"

    @property
    def num_local_experts(self) -> int:
        return int(self.extras.get("num_local_experts", 8))

    @property
    def num_experts_per_tok(self) -> int:
        return int(self.extras.get("num_experts_per_tok", 2))
