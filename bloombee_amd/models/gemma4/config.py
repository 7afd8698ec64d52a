"""Gemma-4 family config (parity: reference models/gemma4/config.py —
text-tower serving of the 2026 Gemma-4 checkpoints).

Key heterogeneity (reference block.py:38-261, server/backend.py:291-306):
  * ``layer_types`` alternates sliding/full attention (5:1 pattern),
  * full-attention layers use ``global_head_dim`` (512 vs 256) and a
    proportional RoPE with partial_rotary_factor 0.25 at theta 1e6; sliding
    layers use full rotary at theta 1e4,
  * optional ``attention_k_eq_v`` (no v_proj on full layers; V = v_norm(k
    projection output), scale-less RMS),
  * optional ``num_kv_shared_layers`` tail layers reusing the last same-type
    layer's KV (no k/v projections at all) — donor and sharer must be
    co-hosted in one server's block range,
  * 4 block norms + q/k norms, all RMS with the (1+weight) convention,
  * attention scaling 1.0 (no 1/sqrt(d); the q/k norms bound magnitudes).

``hidden_size_per_layer_input`` (gemma3n-style per-layer token inputs) is not
implemented; presets set it to 0 and the block asserts.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Tuple

from bloombee_amd.models.base import ModelConfig

GEMMA4_PRESETS = {
    # published default text-tower shape (Gemma4TextConfig defaults)
    "gemma4-9b": dict(
        hidden_size=2304, num_hidden_layers=30, num_attention_heads=8,
        num_key_value_heads=4, head_dim=256, intermediate_size=9216,
        vocab_size=262144, rms_norm_eps=1e-6, tie_word_embeddings=True,
        max_position_embeddings=32768,
        sliding_window=512, global_head_dim=512, sliding_window_pattern=6,
        rope_theta=1000000.0, rope_local_base_freq=10000.0,
        partial_rotary_factor_global=0.25, hidden_size_per_layer_input=0,
    ),
    "gemma4-tiny": dict(
        hidden_size=128, num_hidden_layers=6, num_attention_heads=4,
        num_key_value_heads=2, head_dim=32, intermediate_size=256,
        vocab_size=512, rms_norm_eps=1e-6, tie_word_embeddings=True,
        max_position_embeddings=2048,
        sliding_window=8, global_head_dim=64, sliding_window_pattern=6,
        rope_theta=1000000.0, rope_local_base_freq=10000.0,
        partial_rotary_factor_global=0.25, hidden_size_per_layer_input=0,
    ),
}


@dataclass
class Gemma4Config(ModelConfig):
    model_type: str = "gemma4"

    def layer_type(self, layer: int) -> str:
        explicit = self.extras.get("layer_types")
        if explicit:
            return explicit[layer]
        pattern = int(self.extras.get("sliding_window_pattern", 6))
        return ("full_attention" if (layer + 1) % pattern == 0
                else "sliding_attention")

    def head_dim_for(self, layer: int) -> int:
        if self.layer_type(layer) == "full_attention":
            return int(self.extras.get("global_head_dim", self.head_dim))
        return self.head_dim

    def hkv_for(self, layer: int) -> int:
        if self.layer_type(layer) == "full_attention":
            return int(self.extras.get("num_global_key_value_heads",
                                       self.num_key_value_heads))
        return self.num_key_value_heads

    def rope_for(self, layer: int) -> Tuple[float, float]:
        """-> (theta, partial_rotary_factor)."""
        if self.layer_type(layer) == "full_attention":
            return (self.rope_theta,
                    float(self.extras.get("partial_rotary_factor_global", 0.25)))
        return float(self.extras.get("rope_local_base_freq", 10000.0)), 1.0

    def window_for(self, layer: int) -> int:
        if self.layer_type(layer) == "sliding_attention":
            return int(self.extras.get("sliding_window", 0))
        return 0

    def kv_geometry(self, start: int, end: int):
        layers = range(start, end)
        return ([self.hkv_for(l) for l in layers],
                [self.head_dim_for(l) for l in layers])

    def shared_kv_donor(self, layer: int):
        """Global index of the layer whose KV this layer reuses, or None."""
        nshared = int(self.extras.get("num_kv_shared_layers", 0))
        first_shared = self.num_hidden_layers - nshared
        if nshared <= 0 or layer < first_shared:
            return None
        my_type = self.layer_type(layer)
        for l in range(first_shared - 1, -1, -1):
            if self.layer_type(l) == my_type:
                return l
        return None
