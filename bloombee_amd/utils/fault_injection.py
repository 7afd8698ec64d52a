"""Deterministic RPC fault injection — a test tier the reference lacks
(its failover machinery is exercised only implicitly). With
BBAMD_FAULT_RPC_DROP=<p> (or configure()), each unary RPC attempt fails
with probability p BEFORE reaching the wire; the client retry/re-route
machinery must absorb it. Deterministic: a seeded counter-hash, so a
failing run replays identically."""
from __future__ import annotations

import hashlib
import os
import threading

_lock = threading.Lock()
_drop_p = float(os.environ.get("BBAMD_FAULT_RPC_DROP", "0") or 0)
_seed = int(os.environ.get("BBAMD_FAULT_SEED", "0") or 0)
_counter = 0
injected = 0  # observability for tests


class InjectedRpcFault(ConnectionError):
    pass


_max_faults = None  # stop injecting after this many (None = unlimited)
_methods = None     # restrict injection to these fault points (None = all)


def configure(drop_p: float, seed: int = 0,
              max_faults: "int | None" = None,
              methods=None) -> None:
    global _drop_p, _seed, _counter, injected, _max_faults, _methods
    with _lock:
        _drop_p, _seed, _counter, injected = float(drop_p), seed, 0, 0
        _max_faults = max_faults
        _methods = tuple(methods) if methods else None


def maybe_fail(method: str) -> None:
    """Raise InjectedRpcFault for this attempt with the configured
    probability (never for rpc_info so liveness probes stay truthful)."""
    global _counter, injected
    if _drop_p <= 0.0 or method == "rpc_info":
        return
    if _methods is not None and method not in _methods:
        return
    with _lock:
        if _max_faults is not None and injected >= _max_faults:
            return
        _counter += 1
        h = hashlib.sha256(f"{_seed}:{_counter}".encode()).digest()
        r = int.from_bytes(h[:8], "little") / 2 ** 64
        if r < _drop_p:
            injected += 1
            raise InjectedRpcFault(
                f"injected fault on {method} (attempt {_counter})")
