"""Singleton asyncio loop thread for client-side RPC (the reference relies on
hivemind's RemoteExpertWorker for the same job, client/inference_session.py:330)."""
from __future__ import annotations

import asyncio
import threading
from typing import Any, Coroutine, Dict, Tuple

from bloombee_amd.net.rpc import RpcClient

_lock = threading.Lock()
_loop: asyncio.AbstractEventLoop | None = None
_clients: Dict[Tuple[str, int], RpcClient] = {}


def get_loop() -> asyncio.AbstractEventLoop:
    global _loop
    with _lock:
        if _loop is None or _loop.is_closed():
            loop = asyncio.new_event_loop()
            t = threading.Thread(target=loop.run_forever, daemon=True,
                                 name="bbamd.client-worker")
            t.start()
            _loop = loop
        return _loop


def run_coroutine(coro: Coroutine, timeout: float | None = None) -> Any:
    fut = asyncio.run_coroutine_threadsafe(coro, get_loop())
    return fut.result(timeout)


def get_client(host: str, port: int) -> RpcClient:
    """Shared per-endpoint RpcClient living on the worker loop."""
    key = (host, int(port))
    with _lock:
        if key not in _clients:
            _clients[key] = RpcClient(host, int(port))
        return _clients[key]


def shutdown_clients() -> None:
    """Close every cached client's connection and reader task (quiet
    interpreter shutdown — otherwise the loop is GC'd with pending
    _read_loop tasks and asyncio logs 'Task was destroyed')."""
    with _lock:
        clients, loop = list(_clients.values()), _loop
        _clients.clear()
    if loop is None or loop.is_closed():
        return
    for c in clients:
        try:
            asyncio.run_coroutine_threadsafe(c.close(), loop).result(2)
        except Exception:
            pass


import atexit  # noqa: E402

atexit.register(shutdown_clients)
