"""Memory reporting helpers (parity: reference utils/memory_usage.py —
see_memory_usage / KV-allocation logging, HIP-native via torch.cuda on ROCm)."""
from __future__ import annotations

import torch

from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


def see_memory_usage(message: str = "", force_log: bool = True) -> dict:
    stats = {}
    if torch.cuda.is_available():
        dev = torch.cuda.current_device()
        stats = {
            "allocated_gb": torch.cuda.memory_allocated(dev) / 1e9,
            "reserved_gb": torch.cuda.memory_reserved(dev) / 1e9,
            "max_allocated_gb": torch.cuda.max_memory_allocated(dev) / 1e9,
            "free_gb": torch.cuda.mem_get_info(dev)[0] / 1e9,
            "total_gb": torch.cuda.mem_get_info(dev)[1] / 1e9,
        }
    try:
        import psutil

        vm = psutil.virtual_memory()
        stats["host_used_gb"] = (vm.total - vm.available) / 1e9
        stats["host_total_gb"] = vm.total / 1e9
    except ImportError:
        pass
    if force_log:
        logger.info("%s %s", message,
                    " ".join(f"{k}={v:.2f}" for k, v in stats.items()))
    return stats


def log_kv_alloc(pool, message: str = "kv") -> None:
    logger.info("%s: tokens_left=%d / %d (pages %d x %d)", message,
                pool.tokens_left, pool.max_tokens, pool.n_pages, pool.page_size)
