"""LLaMA family config + published architecture presets (random-init shapes).

Parity target: reference models/llama/config.py (DistributedLlamaConfig).
"""
from __future__ import annotations

from dataclasses import dataclass

from bloombee_amd.models.base import ModelConfig

LLAMA_PRESETS = {
    # BASELINE.json config 2/5 flagship
    "meta-llama/meta-llama-3-8b": dict(
        hidden_size=4096, num_hidden_layers=32, num_attention_heads=32,
        num_key_value_heads=8, intermediate_size=14336, vocab_size=128256,
        max_position_embeddings=8192, rope_theta=500000.0,
    ),
    "llama-3-8b": dict(
        hidden_size=4096, num_hidden_layers=32, num_attention_heads=32,
        num_key_value_heads=8, intermediate_size=14336, vocab_size=128256,
        max_position_embeddings=8192, rope_theta=500000.0,
    ),
    # BASELINE.json config 3 (host-offload target)
    "llama-2-70b": dict(
        hidden_size=8192, num_hidden_layers=80, num_attention_heads=64,
        num_key_value_heads=8, intermediate_size=28672, vocab_size=32000,
        max_position_embeddings=4096, rope_theta=10000.0,
    ),
    "llama-2-7b": dict(
        hidden_size=4096, num_hidden_layers=32, num_attention_heads=32,
        num_key_value_heads=32, intermediate_size=11008, vocab_size=32000,
        max_position_embeddings=4096, rope_theta=10000.0,
    ),
    # small shapes for tests
    "llama-tiny": dict(
        hidden_size=256, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, intermediate_size=512, vocab_size=1024,
        max_position_embeddings=2048, rope_theta=10000.0,
    ),
    "llama-tiny-8l": dict(
        hidden_size=256, num_hidden_layers=8, num_attention_heads=4,
        num_key_value_heads=2, intermediate_size=512, vocab_size=1024,
        max_position_embeddings=2048, rope_theta=10000.0,
    ),
    "llama-mini-gpu": dict(
        hidden_size=1024, num_hidden_layers=4, num_attention_heads=8,
        num_key_value_heads=2, intermediate_size=2816, vocab_size=32000,
        max_position_embeddings=4096, rope_theta=10000.0,
    ),
    # trained speculative-decoding pair (benchmarks/spec_trained.py): both
    # learn the same synthetic grammar so draft/target agreement reflects
    # model quality, not random-init luck
    "llama-spec-target": dict(
        hidden_size=2048, num_hidden_layers=8, num_attention_heads=16,
        num_key_value_heads=4, intermediate_size=5632, vocab_size=8192,
        max_position_embeddings=2048, rope_theta=10000.0,
    ),
    "llama-spec-draft": dict(
        hidden_size=512, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, intermediate_size=1408, vocab_size=8192,
        max_position_embeddings=2048, rope_theta=10000.0,
    ),
}


@dataclass
class LlamaConfig(ModelConfig):
    model_type: str = "llama"
