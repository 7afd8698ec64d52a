"""RTT measurement over the swarm (parity: reference utils/ping.py:31-83
PingAggregator — parallel rpc_info round-trips feeding routing costs and the
next_pings gossip)."""
from __future__ import annotations

import asyncio
import math
import time
from typing import Dict, List, Sequence, Tuple

from bloombee_amd.client.worker import get_client, run_coroutine
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


class PingAggregator:
    def __init__(self, timeout: float = 3.0, ema: float = 0.5):
        self.timeout = timeout
        self.ema = ema
        self.rtts: Dict[Tuple[str, int], float] = {}

    def ping_many(self, endpoints: Sequence[Tuple[str, int]]) -> Dict[Tuple[str, int], float]:
        async def one(ep):
            t0 = time.monotonic()
            try:
                await get_client(*ep).call("rpc_info", {}, timeout=self.timeout)
                return ep, time.monotonic() - t0
            except Exception:
                return ep, math.inf

        async def all_():
            return await asyncio.gather(*[one(tuple(ep)) for ep in endpoints])

        for ep, rtt in run_coroutine(all_(), self.timeout + 5):
            prev = self.rtts.get(ep)
            self.rtts[ep] = rtt if prev is None or math.isinf(prev) else \
                self.ema * rtt + (1 - self.ema) * prev
        return dict(self.rtts)

    def to_dict(self) -> Dict[str, float]:
        return {f"{h}:{p}": v for (h, p), v in self.rtts.items()
                if math.isfinite(v)}
