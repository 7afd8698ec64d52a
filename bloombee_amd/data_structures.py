"""Shared swarm data structures (parity: reference src/bloombee/data_structures.py
and utils/dht.py:74-153).

ModuleUID convention: ``{model_name}{UID_DELIMITER}{block_index}`` — e.g.
``llama-3-8b-bbamd.4`` — same dotted scheme as the reference.
"""
from __future__ import annotations

import dataclasses
import enum
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence, Tuple

UID_DELIMITER = "."
PUBLIC_INITIAL_PEERS: list = []  # no public swarm in this offline environment


class ServerState(enum.IntEnum):
    OFFLINE = 0
    JOINING = 1
    ONLINE = 2


@dataclass
class ServerInfo:
    """Gossiped per-server record (ref data_structures.py:57-93): routing
    inputs (throughput, cache room, next-server pings) ride the DHT."""

    state: ServerState = ServerState.ONLINE
    host: str = "127.0.0.1"
    port: int = 0                      # RPC endpoint for block calls
    throughput: float = 1.0            # blocks/sec per request (rps)
    inference_rps: Optional[float] = None
    forward_rps: Optional[float] = None
    cache_tokens_left: Optional[int] = None
    network_rps: Optional[float] = None
    next_pings: Dict[str, float] = field(default_factory=dict)  # peer_id -> rtt s
    version: str = "bloombee-amd-0.1"
    torch_dtype: str = "bfloat16"
    public_name: Optional[str] = None
    device: Optional[str] = None       # e.g. "cuda:0" / "cpu"
    start_block: Optional[int] = None
    end_block: Optional[int] = None
    # torch.distributed rank when this server is part of a shared world
    # (one process per GPU on a node): lets peers route activation payloads
    # over the RCCL/xGMI device plane (net/channels.py) instead of TCP
    dist_rank: Optional[int] = None

    def to_dict(self) -> dict:
        d = dataclasses.asdict(self)
        d["state"] = int(self.state)
        return d

    @classmethod
    def from_dict(cls, d: dict) -> "ServerInfo":
        d = dict(d)
        d["state"] = ServerState(d.get("state", 2))
        known = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in d.items() if k in known})


@dataclass
class RemoteModuleInfo:
    """All servers currently announcing one block uid."""

    uid: str
    servers: Dict[str, ServerInfo] = field(default_factory=dict)  # peer_id -> info


@dataclass
class RemoteSpanInfo:
    """A contiguous block range hosted by one server (ref compute_spans)."""

    peer_id: str
    start: int
    end: int
    server_info: ServerInfo

    @property
    def length(self) -> int:
        return self.end - self.start


def module_uids(model_name: str, num_blocks: int) -> List[str]:
    return [f"{model_name}{UID_DELIMITER}{i}" for i in range(num_blocks)]


def declare_active_modules(dht, uids: Sequence[str], peer_id: str,
                           server_info: ServerInfo, expiration: float) -> None:
    """Announce this server's blocks (ref utils/dht.py declare_active_modules:
    one DHT record per block uid, subkey = peer id)."""
    info = server_info.to_dict()
    for uid in uids:
        dht.store(uid, peer_id, info, expiration)


def get_remote_module_infos(dht, uids: Sequence[str]) -> List[RemoteModuleInfo]:
    out = []
    now = time.time()
    for uid in uids:
        entry = dht.get(uid)
        servers = {}
        for peer_id, (value, exp) in entry.items():
            if exp > now:
                try:
                    servers[peer_id] = ServerInfo.from_dict(value)
                except Exception:
                    continue
        out.append(RemoteModuleInfo(uid=uid, servers=servers))
    return out


def compute_spans(infos: Sequence[RemoteModuleInfo],
                  min_state: ServerState = ServerState.ONLINE,
                  ) -> Dict[str, RemoteSpanInfo]:
    """Group per-block announcements into contiguous per-server spans
    (ref utils/dht.py compute_spans / client/routing/sequence_info.py)."""
    spans: Dict[str, RemoteSpanInfo] = {}
    for block_idx, info in enumerate(infos):
        for peer_id, server in info.servers.items():
            if server.state < min_state:
                continue
            if peer_id in spans and spans[peer_id].end == block_idx:
                spans[peer_id].end = block_idx + 1
            elif peer_id not in spans:
                spans[peer_id] = RemoteSpanInfo(peer_id, block_idx,
                                                block_idx + 1, server)
            # non-contiguous extra ranges: keep the first span (ref behavior)
    return spans
