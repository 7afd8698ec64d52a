"""rocprof-compatible tracing markers (parity: reference per-step stage
telemetry, SURVEY §5 — log-based [TIMING_SUMMARY] counters plus, here,
roctx ranges that rocprofv3's marker trace can correlate with kernel
dispatches; torch.cuda.nvtx maps to roctx on ROCm).

Enabled with BBAMD_TRACE=1; zero overhead otherwise.
"""
from __future__ import annotations

import contextlib
import os

import torch

_ENABLED = os.environ.get("BBAMD_TRACE", "") not in ("", "0", "false")


def enabled() -> bool:
    return _ENABLED


@contextlib.contextmanager
def trace_range(name: str):
    """roctx range around a stage ('prefill', 'decode', 'push', ...)."""
    if not _ENABLED:
        yield
        return
    torch.cuda.nvtx.range_push(name)
    try:
        yield
    finally:
        torch.cuda.nvtx.range_pop()


def mark(name: str) -> None:
    if _ENABLED:
        torch.cuda.nvtx.mark(name)
