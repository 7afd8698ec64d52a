"""Per-step stage telemetry (parity: reference [TIMING_SUMMARY] /
[TIMING_TABLE] pipeline decomposition, handler.py:1276-1605 — per-session
nic2cpu/compute/cpu2nic stage times aggregated and emitted on close).

MI-native: spans are monotonic host windows; GPU work is bracketed by the
caller (the backend's pool already serializes GPU calls), so host wall time
of the compute span IS the device time plus queueing. Structured events
rather than log-line scraping; `summary()` renders the reference-style
table."""
from __future__ import annotations

import contextlib
import time
from collections import defaultdict
from typing import Dict, List, Optional


class StageTimes:
    """Accumulates named stage durations across steps."""

    def __init__(self):
        self.totals: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)
        self.steps = 0

    @contextlib.contextmanager
    def span(self, name: str):
        t0 = time.monotonic()
        try:
            yield
        finally:
            dt = time.monotonic() - t0
            self.totals[name] += dt
            self.counts[name] += 1

    def bump_step(self):
        self.steps += 1

    def summary(self) -> dict:
        out = {"steps": self.steps}
        for name, tot in sorted(self.totals.items()):
            n = max(1, self.counts[name])
            out[name] = {"total_s": round(tot, 6),
                         "mean_ms": round(1000 * tot / n, 3),
                         "count": self.counts[name]}
        return out

    def table(self) -> str:
        """Reference-style [TIMING_SUMMARY] block."""
        lines = [f"[TIMING_SUMMARY] steps={self.steps}"]
        for name, tot in sorted(self.totals.items(), key=lambda kv: -kv[1]):
            n = max(1, self.counts[name])
            lines.append(f"  {name:<16} total {tot*1000:9.1f} ms   "
                         f"mean {1000*tot/n:7.3f} ms   n={self.counts[name]}")
        return "\n".join(lines)
