"""Qwen3 family config + presets (parity: reference models/qwen3/)."""
from __future__ import annotations

from dataclasses import dataclass

from bloombee_amd.models.base import ModelConfig

QWEN3_PRESETS = {
    "qwen/qwen3-8b": dict(
        hidden_size=4096, num_hidden_layers=36, num_attention_heads=32,
        num_key_value_heads=8, head_dim=128, intermediate_size=12288,
        vocab_size=151936, rms_norm_eps=1e-6, rope_theta=1000000.0,
        max_position_embeddings=32768,
    ),
    "qwen3-8b": dict(
        hidden_size=4096, num_hidden_layers=36, num_attention_heads=32,
        num_key_value_heads=8, head_dim=128, intermediate_size=12288,
        vocab_size=151936, rms_norm_eps=1e-6, rope_theta=1000000.0,
        max_position_embeddings=32768,
    ),
    "qwen3-0.6b": dict(
        hidden_size=1024, num_hidden_layers=28, num_attention_heads=16,
        num_key_value_heads=8, head_dim=128, intermediate_size=3072,
        vocab_size=151936, rms_norm_eps=1e-6, rope_theta=1000000.0,
        tie_word_embeddings=True, max_position_embeddings=32768,
    ),
    "qwen3-tiny": dict(
        hidden_size=256, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, head_dim=64, intermediate_size=512,
        vocab_size=1024, rms_norm_eps=1e-6, max_position_embeddings=2048,
    ),
}


@dataclass
class Qwen3Config(ModelConfig):
    model_type: str = "qwen3"
