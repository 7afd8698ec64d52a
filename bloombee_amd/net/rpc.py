"""Asyncio RPC: unary calls and bidirectional streams over TCP.

Replaces the reference's libp2p/hivemind stub layer (ExpertRequest unary and
streaming RPCs, server/handler.py:798+, client/inference_session.py:85-99)
with a self-contained transport:

  wire frame: [u32 len][frame bytes]   (frame = net/tensors.pack_frame)
  frame.meta carries the envelope: {"id", "kind", "method", ...user meta}
    kind: "req" | "resp" | "err" | "item" | "end"

Unary: req -> resp/err. Streams: the caller opens with kind="req" and
stream=True; both sides then exchange "item" frames under the same id until
"end". One TCP connection multiplexes any number of concurrent calls —
a per-peer connection is kept alive and reused (the reference keeps one
libp2p stream per session; here sessions are ids on a shared connection).
"""
from __future__ import annotations

import asyncio
import itertools
import struct
from typing import Any, AsyncIterator, Awaitable, Callable, Dict, List, Optional, Tuple

import torch

from bloombee_amd.net.tensors import (PendingTensor, pack_frame,
                                      resolve_frame_tensors, unpack_frame)
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)

MAX_FRAME = 1 << 31


class RpcError(RuntimeError):
    pass


async def _read_frame(reader: asyncio.StreamReader) -> bytes:
    head = await reader.readexactly(4)
    (n,) = struct.unpack("<I", head)
    if n > MAX_FRAME:
        raise RpcError(f"frame too large: {n}")
    return await reader.readexactly(n)


def _write_frame(writer: asyncio.StreamWriter, buf: bytes) -> None:
    writer.write(struct.pack("<I", len(buf)) + buf)


class Stream:
    """One end of a bidirectional stream.

    dist_rank: when set, tensors sent on this stream ride the device data
    plane to that rank (RCCL over xGMI for same-node peers) — negotiated at
    stream open (client learns the server's rank from routing, the server
    learns the client's from the open metadata's "drank")."""

    def __init__(self, conn: "_Conn", call_id: int):
        self._conn = conn
        self.call_id = call_id
        self._rx: asyncio.Queue = asyncio.Queue()
        self.closed = False
        self.dist_rank: Optional[int] = None

    async def send(self, meta: dict, tensors: Optional[List[torch.Tensor]] = None,
                   codec: str = "raw") -> None:
        await self._conn.send_env({"id": self.call_id, "kind": "item", **meta},
                                  tensors, codec, dist_rank=self.dist_rank)

    async def send_end(self, meta: Optional[dict] = None) -> None:
        await self._conn.send_env({"id": self.call_id, "kind": "end",
                                   **(meta or {})}, None)

    async def recv(self) -> Optional[Tuple[dict, List[torch.Tensor]]]:
        """None on end-of-stream; raises RpcError on remote error."""
        item = await self._rx.get()
        if item is None:
            return None
        meta, tensors = item
        if meta.get("kind") == "err":
            raise RpcError(meta.get("error", "remote error"))
        if meta.get("kind") == "end":
            return None
        return meta, tensors

    def __aiter__(self) -> AsyncIterator[Tuple[dict, List[torch.Tensor]]]:
        return self._aiter()

    async def _aiter(self):
        while True:
            item = await self.recv()
            if item is None:
                return
            yield item

    def _feed(self, item) -> None:
        self._rx.put_nowait(item)


class _Conn:
    """Shared bidirectional connection state (client or server side)."""

    def __init__(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        self.reader, self.writer = reader, writer
        self.pending: Dict[int, asyncio.Future] = {}
        self.streams: Dict[int, Stream] = {}
        self._wlock = asyncio.Lock()
        self.closed = asyncio.Event()

    async def send_env(self, meta: dict, tensors=None, codec: str = "raw",
                       dist_rank: Optional[int] = None) -> None:
        # pack_frame POSTS the device sends (when dist_rank is set) before
        # the descriptor frame hits the wire, so the receiver's matching
        # irecv is never the first op of the pair
        buf = pack_frame(meta, tensors, codec, dist_rank=dist_rank)
        async with self._wlock:
            _write_frame(self.writer, buf)
            await self.writer.drain()

    def dispatch(self, meta: dict, tensors) -> bool:
        """Route an incoming frame to a pending call/stream. True if routed."""
        cid = meta.get("id")
        kind = meta.get("kind")
        if cid in self.streams and kind in ("item", "end", "err"):
            stream = self.streams[cid]
            stream._feed((meta, tensors))
            if kind in ("end", "err"):
                if kind == "err":
                    stream._feed(None)
                # unregister: no further frames arrive under this id, and
                # long-lived shared per-peer connections must not accumulate
                # one Stream + queue per finished session
                self.streams.pop(cid, None)
            return True
        if cid in self.pending and kind in ("resp", "err"):
            fut = self.pending.pop(cid)
            if not fut.done():
                if kind == "err":
                    fut.set_exception(RpcError(meta.get("error", "remote error")))
                else:
                    fut.set_result((meta, tensors))
            return True
        return False

    def abort_all(self):
        for fut in self.pending.values():
            if not fut.done():
                fut.set_exception(RpcError("connection closed"))
        self.pending.clear()
        for s in self.streams.values():
            s._feed((
                {"kind": "err", "error": "connection closed", "id": s.call_id}, []))
            s._feed(None)
        self.streams.clear()
        self.closed.set()


# unary handler: async (meta, tensors) -> (meta, tensors)
UnaryHandler = Callable[[dict, List[torch.Tensor]], Awaitable[Tuple[dict, List[torch.Tensor]]]]
# stream handler: async (meta, tensors, stream) -> None; owns the stream
StreamHandler = Callable[[dict, List[torch.Tensor], Stream], Awaitable[None]]


class RpcServer:
    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self.host, self.port = host, port
        self._unary: Dict[str, UnaryHandler] = {}
        self._stream: Dict[str, StreamHandler] = {}
        self._server: Optional[asyncio.AbstractServer] = None
        self._conns: set = set()

    def register(self, method: str, fn: UnaryHandler) -> None:
        self._unary[method] = fn

    def register_stream(self, method: str, fn: StreamHandler) -> None:
        self._stream[method] = fn

    async def start(self) -> Tuple[str, int]:
        self._server = await asyncio.start_server(self._on_conn, self.host, self.port)
        sock = self._server.sockets[0]
        self.host, self.port = sock.getsockname()[:2]
        return self.host, self.port

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
        for conn in list(self._conns):
            conn.writer.close()
            try:
                await asyncio.wait_for(conn.writer.wait_closed(), 1)
            except Exception:
                pass

    async def _on_conn(self, reader, writer):
        conn = _Conn(reader, writer)
        self._conns.add(conn)
        try:
            while True:
                buf = await _read_frame(reader)
                meta, tensors = unpack_frame(buf)
                if any(isinstance(t, PendingTensor) for t in tensors):
                    # device-plane payloads: irecvs are already posted (in
                    # frame order); await the data before dispatch so
                    # handlers only ever see real tensors
                    tensors = await resolve_frame_tensors(tensors)
                if conn.dispatch(meta, tensors):
                    continue
                if meta.get("kind") == "req":
                    if meta.get("stream"):
                        # register BEFORE yielding to the handler task: items
                        # may arrive on this connection before it first runs
                        stream = Stream(conn, meta["id"])
                        conn.streams[meta["id"]] = stream
                        asyncio.ensure_future(
                            self._handle_stream(conn, meta, tensors, stream))
                    else:
                        asyncio.ensure_future(self._handle(conn, meta, tensors))
        except (asyncio.IncompleteReadError, ConnectionError, OSError):
            pass
        finally:
            conn.abort_all()
            self._conns.discard(conn)
            writer.close()

    async def _handle(self, conn: _Conn, meta: dict, tensors) -> None:
        cid = meta["id"]
        method = meta.get("method", "")
        try:
            fn = self._unary.get(method)
            if fn is None:
                raise RpcError(f"no method {method!r}")
            out_meta, out_tensors = await fn(meta, tensors)
            # "drank" in the request = the caller's dist rank: response
            # tensors ride the device plane back (set only by callers that
            # have channels enabled)
            await conn.send_env({"id": cid, "kind": "resp", **out_meta},
                                out_tensors, meta.get("codec", "raw"),
                                dist_rank=meta.get("drank"))
        except Exception as e:  # noqa: BLE001 — reported to the peer
            logger.debug("rpc handler error in %s: %s", method, e)
            try:
                await conn.send_env({"id": cid, "kind": "err", "error": str(e)})
            except Exception:
                pass

    async def _handle_stream(self, conn: _Conn, meta: dict, tensors,
                             stream: Stream) -> None:
        cid = meta["id"]
        method = meta.get("method", "")
        try:
            fn = self._stream.get(method)
            if fn is None:
                raise RpcError(f"no stream method {method!r}")
            await fn(meta, tensors, stream)
        except Exception as e:  # noqa: BLE001 — reported to the peer
            logger.debug("rpc stream handler error in %s: %s", method, e)
            try:
                await conn.send_env({"id": cid, "kind": "err", "error": str(e)})
            except Exception:
                pass
        finally:
            conn.streams.pop(cid, None)


class RpcClient:
    """Per-peer client with a lazily (re)established multiplexed connection."""

    _ids = itertools.count(1)

    def __init__(self, host: str, port: int):
        self.host, self.port = host, port
        self._conn: Optional[_Conn] = None
        self._reader_task: Optional[asyncio.Task] = None
        self._lock = asyncio.Lock()

    async def _ensure(self) -> _Conn:
        async with self._lock:
            if self._conn is not None and not self._conn.closed.is_set():
                return self._conn
            reader, writer = await asyncio.open_connection(self.host, self.port)
            self._conn = _Conn(reader, writer)
            self._reader_task = asyncio.ensure_future(self._read_loop(self._conn))
            return self._conn

    async def _read_loop(self, conn: _Conn):
        try:
            while True:
                buf = await _read_frame(conn.reader)
                meta, tensors = unpack_frame(buf)
                if any(isinstance(t, PendingTensor) for t in tensors):
                    tensors = await resolve_frame_tensors(tensors)
                conn.dispatch(meta, tensors)
        except (asyncio.IncompleteReadError, ConnectionError, OSError):
            pass
        finally:
            conn.abort_all()

    async def call(self, method: str, meta: Optional[dict] = None,
                   tensors: Optional[List[torch.Tensor]] = None,
                   codec: str = "raw", timeout: Optional[float] = 30.0,
                   dist_rank: Optional[int] = None,
                   ) -> Tuple[dict, List[torch.Tensor]]:
        from bloombee_amd.utils.fault_injection import maybe_fail
        maybe_fail(method)  # no-op unless BBAMD_FAULT_RPC_DROP is set
        conn = await self._ensure()
        cid = next(self._ids)
        fut: asyncio.Future = asyncio.get_event_loop().create_future()
        conn.pending[cid] = fut
        await conn.send_env({"id": cid, "kind": "req", "method": method,
                             "codec": codec, **(meta or {})}, tensors, codec,
                            dist_rank=dist_rank)
        return await asyncio.wait_for(fut, timeout)

    async def open_stream(self, method: str, meta: Optional[dict] = None,
                          tensors: Optional[List[torch.Tensor]] = None,
                          codec: str = "raw",
                          dist_rank: Optional[int] = None) -> Stream:
        conn = await self._ensure()
        cid = next(self._ids)
        stream = Stream(conn, cid)
        stream.dist_rank = dist_rank
        conn.streams[cid] = stream
        await conn.send_env({"id": cid, "kind": "req", "method": method,
                             "stream": True, "codec": codec, **(meta or {})},
                            tensors, codec, dist_rank=dist_rank)
        return stream

    async def close(self):
        if self._conn is not None:
            self._conn.writer.close()
            self._conn.abort_all()
        if self._reader_task is not None:
            self._reader_task.cancel()
