"""Kademlia-style DHT: the decentralized control plane.

Replaces hivemind's DHT-over-libp2p (the reference's discovery substrate,
SURVEY.md §1 control plane; utils/dht.py:74-153) with a compact Kademlia over
bloombee_amd.net.rpc:

  * 160-bit node ids, XOR metric, k-buckets (k=8), alpha=3 iterative lookups.
  * Values are dictionaries of subkeys with per-subkey expiration — the shape
    hivemind stores module announcements in (key = block uid, subkey =
    peer id, value = server info, expiration) so `declare_active_modules` /
    `get_remote_module_infos` port over directly (server announces every
    update_period; records expire at 2x, ref server/server.py:177-179).
  * Replication: STORE goes to the k closest nodes to the key hash.

`DhtNode` is fully asyncio; `Dht` wraps it in a daemon thread with a
synchronous facade (the reference runs a DHT process; a thread suffices —
no GIL-heavy work here).
"""
from __future__ import annotations

import asyncio
import hashlib
import os
import threading
import time
from typing import Any, Dict, List, Optional, Tuple

from bloombee_amd.net.rpc import RpcClient, RpcError, RpcServer
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)

K_BUCKET = 8
ALPHA = 3
ID_BITS = 160


def _key_hash(key: str) -> int:
    return int.from_bytes(hashlib.sha1(key.encode()).digest(), "big")


def _rand_id() -> int:
    return int.from_bytes(os.urandom(20), "big")


class DhtNode:
    """One DHT participant: routing table + local store + RPC endpoints."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0,
                 node_id: Optional[int] = None):
        self.node_id = node_id if node_id is not None else _rand_id()
        self.server = RpcServer(host, port)
        self.endpoint: Optional[Tuple[str, int]] = None
        # routing table: bucket index -> list of (node_id, host, port)
        self.buckets: List[List[Tuple[int, str, int]]] = [[] for _ in range(ID_BITS)]
        # key_hash -> {subkey: (value, expiration_ts)}
        self.store_: Dict[int, Dict[str, Tuple[Any, float]]] = {}
        self._clients: Dict[Tuple[str, int], RpcClient] = {}

    # ---- lifecycle ------------------------------------------------------
    async def start(self, initial_peers: Optional[List[Tuple[str, int]]] = None):
        for m in ("ping", "store", "find_node", "find_value"):
            self.server.register(f"dht_{m}", getattr(self, f"_rpc_{m}"))
        self.endpoint = await self.server.start()
        for peer in initial_peers or []:
            try:
                await self._ping_peer(tuple(peer))
            except (RpcError, OSError, asyncio.TimeoutError):
                logger.warning("initial peer %s unreachable", peer)
        if any(self.buckets[i] for i in range(ID_BITS)):
            await self.lookup_nodes(self.node_id)  # populate table
        return self.endpoint

    async def stop(self):
        await self.server.stop()
        for c in self._clients.values():
            await c.close()

    # ---- routing table --------------------------------------------------
    def _bucket_of(self, nid: int) -> int:
        d = nid ^ self.node_id
        return d.bit_length() - 1 if d else 0

    def _touch(self, nid: int, host: str, port: int):
        if nid == self.node_id:
            return
        b = self.buckets[self._bucket_of(nid)]
        entry = (nid, host, port)
        for i, (eid, _, _) in enumerate(b):
            if eid == nid:
                b.pop(i)
                break
        b.append(entry)
        if len(b) > K_BUCKET:
            b.pop(0)

    def _closest(self, target: int, n: int = K_BUCKET) -> List[Tuple[int, str, int]]:
        all_nodes = [e for b in self.buckets for e in b]
        all_nodes.sort(key=lambda e: e[0] ^ target)
        return all_nodes[:n]

    def _client(self, host: str, port: int) -> RpcClient:
        key = (host, port)
        if key not in self._clients:
            self._clients[key] = RpcClient(host, port)
        return self._clients[key]

    async def _ping_peer(self, peer: Tuple[str, int]):
        meta, _ = await self._client(*peer).call(
            "dht_ping", {"nid": str(self.node_id), "host": self.endpoint[0],
                         "port": self.endpoint[1]}, timeout=5)
        self._touch(int(meta["nid"]), peer[0], peer[1])

    # ---- RPC handlers ---------------------------------------------------
    def _note_sender(self, meta: dict):
        try:
            self._touch(int(meta["nid"]), meta["host"], int(meta["port"]))
        except (KeyError, ValueError):
            pass

    async def _rpc_ping(self, meta, tensors):
        self._note_sender(meta)
        return {"nid": str(self.node_id)}, []

    async def _rpc_store(self, meta, tensors):
        self._note_sender(meta)
        kh = int(meta["key_hash"])
        sub = self.store_.setdefault(kh, {})
        now = time.time()
        for subkey, value, exp in meta["entries"]:
            if exp > now and (subkey not in sub or sub[subkey][1] <= exp):
                sub[subkey] = (value, exp)
        return {"ok": True}, []

    async def _rpc_find_node(self, meta, tensors):
        self._note_sender(meta)
        target = int(meta["target"])
        nodes = [(str(nid), h, p) for nid, h, p in self._closest(target)]
        return {"nodes": nodes}, []

    async def _rpc_find_value(self, meta, tensors):
        self._note_sender(meta)
        kh = int(meta["key_hash"])
        now = time.time()
        found = {}
        if kh in self.store_:
            found = {sk: (v, e) for sk, (v, e) in self.store_[kh].items() if e > now}
        nodes = [(str(nid), h, p) for nid, h, p in self._closest(kh)]
        return {"value": found, "nodes": nodes}, []

    # ---- iterative operations ------------------------------------------
    def _self_meta(self) -> dict:
        return {"nid": str(self.node_id), "host": self.endpoint[0],
                "port": self.endpoint[1]}

    async def lookup_nodes(self, target: int) -> List[Tuple[int, str, int]]:
        """Iterative FIND_NODE: returns k closest live nodes to target."""
        shortlist = {e[0]: e for e in self._closest(target, K_BUCKET)}
        queried = set()
        while True:
            candidates = sorted(shortlist.values(), key=lambda e: e[0] ^ target)
            batch = [e for e in candidates if e[0] not in queried][:ALPHA]
            if not batch:
                return candidates[:K_BUCKET]
            results = await asyncio.gather(
                *[self._q_find_node(e, target) for e in batch],
                return_exceptions=True)
            for e, res in zip(batch, results):
                queried.add(e[0])
                if isinstance(res, Exception):
                    shortlist.pop(e[0], None)
                    continue
                for nid_s, h, p in res:
                    nid = int(nid_s)
                    if nid != self.node_id:
                        shortlist.setdefault(nid, (nid, h, p))

    async def _q_find_node(self, entry, target) -> list:
        meta, _ = await self._client(entry[1], entry[2]).call(
            "dht_find_node", {**self._self_meta(), "target": str(target)}, timeout=5)
        self._touch(*entry)
        return meta["nodes"]

    async def store(self, key: str, subkey: str, value: Any,
                    expiration: float) -> int:
        """Store on the k closest nodes (incl. self if among them). Returns
        the number of successful replicas."""
        kh = _key_hash(key)
        targets = await self.lookup_nodes(kh)
        entries = [(subkey, value, expiration)]
        ok = 0
        # always keep a local replica — a lone bootstrap node has no peers
        await self._rpc_store({**self._self_meta(), "key_hash": str(kh),
                               "entries": entries}, [])
        for nid, h, p in targets:
            try:
                await self._client(h, p).call(
                    "dht_store", {**self._self_meta(), "key_hash": str(kh),
                                  "entries": entries}, timeout=5)
                ok += 1
            except (RpcError, OSError, asyncio.TimeoutError):
                pass
        return ok + 1

    async def get(self, key: str) -> Dict[str, Tuple[Any, float]]:
        """-> {subkey: (value, expiration)} merged over responders."""
        kh = _key_hash(key)
        now = time.time()
        merged: Dict[str, Tuple[Any, float]] = {
            sk: (v, e) for sk, (v, e) in self.store_.get(kh, {}).items() if e > now}
        targets = self._closest(kh, K_BUCKET)
        queried = set()
        for _ in range(3):  # a few waves of alpha queries
            batch = [t for t in targets if t[0] not in queried][:ALPHA]
            if not batch:
                break
            results = await asyncio.gather(
                *[self._q_find_value(e, kh) for e in batch], return_exceptions=True)
            for e, res in zip(batch, results):
                queried.add(e[0])
                if isinstance(res, Exception):
                    continue
                value, nodes = res
                for sk, (v, exp) in value.items():
                    if exp > now and (sk not in merged or merged[sk][1] < exp):
                        merged[sk] = (v, exp)
                for nid_s, h, p in nodes:
                    nid = int(nid_s)
                    if nid != self.node_id and all(nid != t[0] for t in targets):
                        targets.append((nid, h, p))
            targets.sort(key=lambda t: t[0] ^ kh)
        return merged

    async def _q_find_value(self, entry, kh):
        meta, _ = await self._client(entry[1], entry[2]).call(
            "dht_find_value", {**self._self_meta(), "key_hash": str(kh)}, timeout=5)
        self._touch(*entry)
        return meta["value"], meta["nodes"]


class Dht:
    """Thread-hosted DhtNode with a synchronous facade (the reference's `DHT`
    object, hivemind_compat — here a thread instead of a forked process)."""

    def __init__(self, initial_peers: Optional[List[Tuple[str, int]]] = None,
                 host: str = "127.0.0.1", port: int = 0, start: bool = True):
        self.node = DhtNode(host, port)
        self.loop = asyncio.new_event_loop()
        self.initial_peers = [tuple(p) for p in initial_peers or []]
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="bbamd.dht")
        self._ready = threading.Event()
        if start:
            self.run()

    def _run(self):
        asyncio.set_event_loop(self.loop)
        self.loop.run_until_complete(self.node.start(self.initial_peers))
        self._ready.set()
        self.loop.run_forever()

    def run(self):
        self._thread.start()
        if not self._ready.wait(timeout=30):
            raise RuntimeError("DHT failed to start")

    @property
    def endpoint(self) -> Tuple[str, int]:
        return self.node.endpoint

    def _call(self, coro, timeout=30):
        fut = asyncio.run_coroutine_threadsafe(coro, self.loop)
        return fut.result(timeout)

    def store(self, key: str, subkey: str, value: Any, expiration: float) -> int:
        return self._call(self.node.store(key, subkey, value, expiration))

    def get(self, key: str) -> Dict[str, Tuple[Any, float]]:
        return self._call(self.node.get(key))

    def shutdown(self):
        if self._thread.is_alive():
            self._call(self.node.stop())
            self.loop.call_soon_threadsafe(self.loop.stop)
            self._thread.join(timeout=5)
