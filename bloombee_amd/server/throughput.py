"""Device throughput measurement for routing/announcements.

Reference: server/throughput.py:44-198 — measure per-block inference rps
(1 token x N steps) and forward rps (1024 tokens x K steps), cache the
result in a JSON file keyed by (device, dtype, model shape), report
min(forward_rps / ((num_blocks+1)/2), network_rps).
"""
from __future__ import annotations

import json
import os
import time
from pathlib import Path
from typing import Optional

import torch

from bloombee_amd.models.base import ModelConfig
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)

CACHE_FILE = os.path.join(
    os.environ.get("BBAMD_CACHE_DIR",
                   os.path.expanduser("~/.cache/bloombee_amd")),
    "throughput_v1.json")


def measure_compute_rps(config: ModelConfig, device: str = "cpu",
                        n_tokens: int = 1, n_steps: int = 10,
                        batch: int = 1) -> float:
    """Steps/sec of one block at (batch, n_tokens) on `device`."""
    from bloombee_amd.engine import BlockStack

    stack = BlockStack(config, 0, 1, device=device, seed=0)
    kv = stack.make_kv(max(4096, batch * (n_tokens * (n_steps + 2) + 8)))
    h = kv.allocate(batch, n_tokens * (n_steps + 2) + 8)
    hidden = torch.randn(batch, n_tokens, config.hidden_size).to(
        config.dtype).to(device)
    pos = 0
    # warmup
    sp = torch.full((batch,), pos, dtype=torch.int32, device=device)
    h.extend(n_tokens)
    stack.forward_inference(hidden, h, sp)
    pos += n_tokens
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(n_steps):
        sp = torch.full((batch,), pos, dtype=torch.int32, device=device)
        h.extend(n_tokens)
        stack.forward_inference(hidden, h, sp)
        pos += n_tokens
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    dt = time.monotonic() - t0
    h.close()
    return n_steps / dt


def get_server_throughput(config: ModelConfig, device: str, num_blocks: int,
                          force_eval: bool = False,
                          network_rps: Optional[float] = None,
                          relayed: bool = False) -> dict:
    """network_rps: advertised NIC steps/s cap (the reference measures it
    with speedtest-cli — impossible offline, so it is operator-provided via
    --network-rps); reported throughput = min(compute, network*penalty)
    (ref throughput.py:123-133)."""
    key = f"{config.model_type}-{config.hidden_size}-{device.split(':')[0]}-{config.torch_dtype}"
    cache = {}
    p = Path(CACHE_FILE)
    if p.exists() and not force_eval:
        try:
            cache = json.loads(p.read_text())
        except Exception:
            cache = {}
    if key not in cache:
        logger.info("measuring compute throughput for %s ...", key)
        inference_rps = measure_compute_rps(config, device, n_tokens=1, n_steps=5)
        forward_rps = measure_compute_rps(config, device, n_tokens=128, n_steps=3)
        cache[key] = {"inference_rps": inference_rps, "forward_rps": forward_rps}
        try:
            p.parent.mkdir(parents=True, exist_ok=True)
            p.write_text(json.dumps(cache))
        except OSError:
            pass
    ent = cache[key]
    throughput = ent["forward_rps"] / max(1.0, (num_blocks + 1) / 2)
    if network_rps is not None:
        throughput = min(throughput, network_rps * (0.2 if relayed else 1.0))
    return {"throughput": throughput, "network_rps": network_rps, **ent}
