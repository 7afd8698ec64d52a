"""Offload policy (parity: reference flexgen_utils/policy.py:10-55 Policy —
the percent splits and compression switches, re-scoped to the MI355X design:
weights stream host->HBM on a side stream sized against 288 GB HBM3E; the
disk tier backs the host pool via np.memmap)."""
from __future__ import annotations

from dataclasses import dataclass


@dataclass
class OffloadPolicy:
    # fraction of BLOCKS kept fully resident in HBM; the rest live on host
    # (the reference splits per-tensor by percent; block granularity maps
    # better to a double-buffered HBM arena + per-block prefetch)
    weight_gpu_percent: float = 100.0
    weight_disk_percent: float = 0.0     # of the host tier, memmap-backed
    cache_gpu_percent: float = 100.0     # KV pages resident fraction
    pin_weight: bool = True
    compress_weight: bool = False        # 4-bit group quant on the host tier
    compress_cache: bool = False
    # top-k sparse decode attention: fraction of cache positions whose V
    # rows join the weighted sum (ref Policy.attn_sparsity + the
    # _sparse_attention_value path); 1.0 = exact dense
    attn_sparsity: float = 1.0
    overlap: bool = True                 # prefetch block i+1 during block i
    prefetch_depth: int = 1

    @property
    def offloads_weights(self) -> bool:
        return self.weight_gpu_percent < 100.0
