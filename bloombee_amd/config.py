"""Typed runtime configuration with environment overrides.

The reference scatters ~60 ``BLOOMBEE_*`` env switches across
microbatch_config.py / debug_config.py / lossless_transport.py (reference
README.environment-switches.md). Here they are consolidated into one typed
config (SURVEY.md §5 "Config / flag system": "the new framework should
consolidate (3) into a typed config with env overrides").

Env prefix: ``BBAMD_``.
"""
from __future__ import annotations

import dataclasses
import os
from dataclasses import dataclass, field
from typing import Optional


def _parse_mixed(raw: str) -> str:
    raw = raw.strip().lower()
    if raw in ("1", "on", "true", "yes"):
        return "on"
    if raw == "auto":
        return "auto"
    return "off"


def _env(name: str, default, cast=None):
    raw = os.environ.get(f"BBAMD_{name}")
    if raw is None:
        return default
    if cast is None:
        cast = type(default) if default is not None else str
    if cast is bool:
        return raw.lower() not in ("0", "false", "no", "off", "")
    return cast(raw)


@dataclass
class KVConfig:
    """Paged KV cache policy (replaces reference MemoryCache/KVCacheManager/
    PagedKVTable trio, memory_cache.py / memory_cache_manager.py / paged_kv.py —
    designed paged-first, SURVEY.md §7 hard-part 2)."""

    page_size: int = field(default_factory=lambda: _env("KV_PAGE_SIZE", 16))
    # Fraction of free device memory the KV pool may claim when the budget is
    # not set explicitly (the reference sizes from --attn_cache_tokens).
    max_tokens: Optional[int] = field(default_factory=lambda: _env("KV_MAX_TOKENS", None, int))
    alloc_timeout: float = field(default_factory=lambda: _env("KV_ALLOC_TIMEOUT", 60.0))
    # BBAMD_MIXED_ATTN: host-swapped sessions resume DECODING with their
    # committed KV left in host memory (mixed-device attention, exact
    # log-sum-exp merge) instead of being restored to HBM first (ref
    # _mixed_device_attention, pytorch_backend.py:969-1014).
    #   off  (default): always restore (swap_in)
    #   on/1: always decode mixed (capacity mode)
    #   auto: restore when the device pool currently has room for the
    #         session's pages, go mixed only when it does not — mixed-attn
    #         as the measured capacity fallback rather than a blanket mode
    mixed_attn: str = field(default_factory=lambda: _parse_mixed(
        os.environ.get("BBAMD_MIXED_ATTN", "off")))


@dataclass
class MicrobatchConfig:
    """Micro-batch pipelining policy (reference microbatch_config.py:27-123)."""

    enabled: bool = field(default_factory=lambda: _env("MICROBATCH", True))
    micro_batch_size: int = field(default_factory=lambda: _env("MICRO_BATCH_SIZE", 4))
    min_batch_to_split: int = field(default_factory=lambda: _env("MIN_BATCH_TO_SPLIT", 8))
    # BBAMD_KV_MULTIPLEX=1: micro-batched sessions keep only ~2 slices of
    # KV device-resident; the rest cycles through pinned host snapshots on
    # the staging stream while the previous slice computes (ref per-MB KV
    # offload/prefetch, memory_cache_manager.py:944-1371)
    kv_multiplex: bool = field(default_factory=lambda: _env("KV_MULTIPLEX", False))


@dataclass
class CompressionConfig:
    """Lossless wire compression (reference utils/lossless_transport.py)."""

    enabled: bool = field(default_factory=lambda: _env("WIRE_COMPRESSION", False))
    codec: str = field(default_factory=lambda: _env("WIRE_CODEC", "zlib-split"))
    min_size_bytes: int = field(default_factory=lambda: _env("WIRE_MIN_SIZE", 1 << 16))
    min_gain: float = field(default_factory=lambda: _env("WIRE_MIN_GAIN", 0.05))
    level: int = field(default_factory=lambda: _env("WIRE_LEVEL", 1))


@dataclass
class DebugConfig:
    """Hierarchical debug switches (reference utils/debug_config.py)."""

    enabled: bool = field(default_factory=lambda: _env("DEBUG", False))
    kv: bool = field(default_factory=lambda: _env("DEBUG_KV", False))
    microbatch: bool = field(default_factory=lambda: _env("DEBUG_MICROBATCH", False))
    inference: bool = field(default_factory=lambda: _env("DEBUG_INFERENCE", False))
    transport: bool = field(default_factory=lambda: _env("DEBUG_TRANSPORT", False))

    def channel(self, name: str) -> bool:
        return self.enabled or bool(getattr(self, name, False))


@dataclass
class RuntimeConfig:
    """Top-level runtime knobs."""

    kv: KVConfig = field(default_factory=KVConfig)
    microbatch: MicrobatchConfig = field(default_factory=MicrobatchConfig)
    compression: CompressionConfig = field(default_factory=CompressionConfig)
    debug: DebugConfig = field(default_factory=DebugConfig)
    # Capture the per-step decode inner loop in a hipGraph (utils/cuda_graphs.py
    # analog; on by default — MI355X decode is launch-bound without it).
    use_hip_graphs: bool = field(default_factory=lambda: _env("HIP_GRAPHS", True))
    # Step profiling (reference BLOOMBEE_STEP_PROFILE, backend.py:59-60)
    step_profile: bool = field(default_factory=lambda: _env("STEP_PROFILE", False))

    def replace(self, **kw) -> "RuntimeConfig":
        return dataclasses.replace(self, **kw)


_GLOBAL: Optional[RuntimeConfig] = None


def get_config() -> RuntimeConfig:
    global _GLOBAL
    if _GLOBAL is None:
        _GLOBAL = RuntimeConfig()
    return _GLOBAL


def set_config(cfg: RuntimeConfig) -> None:
    global _GLOBAL
    _GLOBAL = cfg
