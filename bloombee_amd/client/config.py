"""Client-side configuration (parity: reference client/config.py:19-42)."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Sequence, Tuple


@dataclass
class ClientConfig:
    initial_peers: Sequence[Tuple[str, int]] = field(default_factory=list)
    dht_port: int = 0
    daemon_startup_timeout: float = 30.0

    show_route: str = "inference"
    routing_mode: str = "max_throughput"   # or "min_latency"
    allowed_servers: Optional[List[str]] = None
    blocked_servers: Optional[List[str]] = None

    request_timeout: float = 30.0
    session_timeout: float = 300.0
    step_timeout: float = 60.0
    max_retries: Optional[int] = 3
    min_backoff: float = 1.0
    max_backoff: float = 60.0
    update_period: float = 30.0
    ban_timeout: float = 15.0

    active_adapter: "Optional[str]" = None  # LoRA adapter name servers apply
    # activation wire codec: "raw" | "zlib" | "bsplit+zlib" (byte-split lanes
    # before DEFLATE — the reference lossless transport's best default,
    # lossless_transport.py:1604-1667)
    wire_codec: str = "raw"
    use_server_to_server: bool = True      # s2s activation push during decode
    push_only_downstream_decode: bool = True
    # keep per-step span-0 inputs for failover history replay (ref
    # inference_session.py:71,139-150). Benchmarks on a healthy single node
    # turn this off: the history pins every step's activations in memory.
    keep_history: bool = True
