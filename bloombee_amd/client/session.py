"""Multi-step decode session across a chain of remote spans.

Parity: reference InferenceSession / _ServerInferenceSession
(client/inference_session.py:511, :206-406) including:
  * one bidirectional rpc_inference stream per span,
  * server→server push: middle spans are opened quiet with a push_to target
    so during decode the client only sends to span 0 and receives from the
    last span (ref push_only_downstream_decode, :178-196, 404-406),
  * failover: on span failure the route is rebuilt and the session history
    (span-0 inputs) is replayed through fresh sessions to rebuild KV
    (ref `history` replay + _update_sequence, :71, 139-150, 802-831).
"""
from __future__ import annotations

import time
import uuid
from typing import List, Optional, Tuple

import torch

from bloombee_amd.client.config import ClientConfig
from bloombee_amd.client.routing import (MissingBlocksError,
                                         RemoteSequenceManager,
                                         RemoteSpanInfo)
from bloombee_amd.client.worker import get_client, run_coroutine
from bloombee_amd.net.rpc import RpcError, Stream
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)

# On Python < 3.11, concurrent.futures.TimeoutError and asyncio.TimeoutError
# are NOT the builtin TimeoutError — a step timeout must still trigger
# failover (found by fault injection on the s2s push path)
import asyncio as _asyncio  # noqa: E402
import concurrent.futures as _cf  # noqa: E402

_RETRYABLE = (OSError, TimeoutError, ConnectionError,
              _cf.TimeoutError, _asyncio.TimeoutError)


class _SpanSession:
    """Client end of one span's rpc_inference stream."""

    def __init__(self, span: RemoteSpanInfo, stream: Stream, session_id: str,
                 quiet: bool, codec: str = "raw"):
        self.span = span
        self.stream = stream
        self.session_id = session_id
        self.quiet = quiet
        self.codec = codec

    @classmethod
    def create(cls, span: RemoteSpanInfo, batch_size: int, max_length: int,
               push_to: Optional[Tuple] = None,
               push_only_recv: bool = False, quiet: bool = False,
               timeout: float = 30.0, adapter: Optional[str] = None,
               codec: str = "raw") -> "_SpanSession":
        client = get_client(span.server_info.host, span.server_info.port)
        sid = uuid.uuid4().hex
        # device data plane: if this server shares our torch.distributed
        # world, step payloads ride RCCL/xGMI to its rank; "drank" tells it
        # where to send replies (net/channels.py)
        from bloombee_amd.net.channels import channels
        span_rank = (span.server_info.dist_rank if channels.enabled else None)

        async def open_():
            stream = await client.open_stream("rpc_inference", {
                "session_id": sid,
                "max_length": max_length,
                "batch_size": batch_size,
                "push_to": list(push_to) if push_to else None,
                "push_only_recv": push_only_recv,
                "quiet": quiet,
                "adapter": adapter,
                "codec": codec,
                "drank": channels.rank,
            }, dist_rank=span_rank)
            first = await stream.recv()
            if first is None or not first[0].get("ok"):
                raise RpcError(f"failed to open session on {span.peer_id}")
            return stream

        from bloombee_amd.utils.fault_injection import maybe_fail
        maybe_fail("open_inference")
        stream = run_coroutine(open_(), timeout)
        return cls(span, stream, sid, quiet, codec)

    def step(self, hidden: torch.Tensor, pos: int, step: int,
             timeout: float,
             prompts: Optional[torch.Tensor] = None) -> Optional[torch.Tensor]:
        from bloombee_amd.utils.fault_injection import maybe_fail
        maybe_fail("inference_item")  # no-op unless fault injection is armed

        async def go():
            payload = [hidden] if prompts is None else [hidden, prompts]
            await self.stream.send({"pos": pos, "step": step}, payload,
                                   codec=self.codec)
            item = await self.stream.recv()
            if item is None:
                raise RpcError(f"stream closed by {self.span.peer_id}")
            meta, tensors = item
            return tensors[0] if tensors else None

        return run_coroutine(go(), timeout)

    def send_only(self, hidden: torch.Tensor, pos: int, step: int,
                  timeout: float,
                  prompts: Optional[torch.Tensor] = None) -> None:
        async def go():
            payload = [hidden] if prompts is None else [hidden, prompts]
            await self.stream.send({"pos": pos, "step": step}, payload,
                                   codec=self.codec)

        run_coroutine(go(), timeout)

    def recv_only(self, timeout: float) -> Tuple[dict, List[torch.Tensor]]:
        async def go():
            item = await self.stream.recv()
            if item is None:
                raise RpcError(f"stream closed by {self.span.peer_id}")
            return item

        return run_coroutine(go(), timeout)

    def close(self):
        async def go():
            try:
                await self.stream.send({"close": True})
                await self.stream.send_end()
            except Exception:
                pass

        try:
            run_coroutine(go(), 5)
        except Exception:
            pass


class InferenceSession:
    def __init__(self, manager: RemoteSequenceManager, max_length: int,
                 config: Optional[ClientConfig] = None,
                 allow_push: bool = True):
        self.manager = manager
        self.config = config or manager.config
        self.allow_push = allow_push
        self.max_length = max_length
        self.batch_size: Optional[int] = None
        self.spans: List[_SpanSession] = []
        self.position = 0           # committed tokens across the chain
        self.step_count = 0
        # (pos, span0 input, deep prompts or None)
        self.history: List[Tuple[int, torch.Tensor, Optional[torch.Tensor]]] = []
        self._closed = False

    # -- chain management -------------------------------------------------
    def _open_chain(self, batch_size: int, replay: bool,
                    allow_push: Optional[bool] = None) -> None:
        if allow_push is not None:
            self.allow_push = self.allow_push and allow_push
        route = self.manager.make_sequence(
            0, self.manager.num_blocks,
            cache_tokens_needed=batch_size * self.max_length)
        use_push = (self.allow_push and self.config.use_server_to_server
                    and len(route) > 1)
        sessions: List[Optional[_SpanSession]] = [None] * len(route)
        # open back-to-front so each span knows its downstream session id
        push_to = None
        for i in reversed(range(len(route))):
            quiet = use_push and i < len(route) - 1
            push_only_recv = use_push and i > 0
            s = _SpanSession.create(
                route[i], batch_size, self.max_length,
                push_to=push_to, push_only_recv=push_only_recv, quiet=quiet,
                timeout=self.config.request_timeout,
                adapter=getattr(self.config, "active_adapter", None),
                codec=getattr(self.config, "wire_codec", "raw"))
            sessions[i] = s
            push_to = (route[i].server_info.host, route[i].server_info.port,
                       s.session_id,
                       route[i].server_info.dist_rank) if use_push else None
        self.spans = sessions  # type: ignore[assignment]
        self._push_mode = use_push
        if replay and self.history:
            logger.info("replaying %d cached steps into the new chain",
                        len(self.history))
            for pos, hidden, pr in self.history:
                self._chain_step(hidden, pos, replay=True, prompts=pr)

    def _chain_step(self, hidden: torch.Tensor, pos: int,
                    replay: bool = False,
                    prompts: Optional[torch.Tensor] = None) -> torch.Tensor:
        t = self.config.step_timeout

        def _slice(span):
            # deep p-tune: full-depth (num_blocks, pre, H) -> this span's rows
            return (prompts[span.start:span.end]
                    if prompts is not None else None)

        if self._push_mode:
            first, last = self.spans[0], self.spans[-1]
            if len(self.spans) == 1:
                return first.step(hidden, pos, self.step_count, t,
                                  prompts=_slice(first.span))
            # push mode with deep prompts would need per-hop prompt routing;
            # sessions with prompts are opened pushless by step() below
            first.send_only(hidden, pos, self.step_count, t,
                            prompts=_slice(first.span))
            # quiet middle spans send nothing; output arrives from the last
            meta, tensors = last.recv_only(t)
            return tensors[0]
        out = hidden
        for s in self.spans:
            out = s.step(out, pos, self.step_count, t, prompts=_slice(s.span))
        return out

    def _ban_dead_spans(self) -> None:
        """Probe each span's endpoint; ban the ones that don't answer
        (failure attribution before re-routing, ref sequence_manager bans)."""
        for s in self.spans:
            try:
                client = get_client(s.span.server_info.host,
                                    s.span.server_info.port)
                run_coroutine(client.call("rpc_info", {}, timeout=3), 5)
                self.manager.on_request_success(s.span.peer_id)
            except Exception:
                self.manager.on_request_failure(s.span.peer_id)

    # -- public API -------------------------------------------------------
    def step(self, hidden: torch.Tensor,
             prompts: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Run the full remote chain on `hidden` (B, T, H) at the current
        position; returns the last span's output hidden states. prompts:
        optional full-depth deep p-tune tensor (num_blocks, pre, H); each
        span receives its block-range slice with the step item."""
        if self._closed:
            raise RuntimeError("session is closed")
        first_open = self.batch_size is None
        if first_open:
            self.batch_size = hidden.shape[0]
        pos = self.position
        attempt = 0
        while True:
            try:
                # chain (re)open lives INSIDE the retry loop: a failure while
                # establishing span streams must re-route like a step failure
                if first_open or not self.spans:
                    self._open_chain(self.batch_size,
                                     replay=not first_open and bool(self.history),
                                     allow_push=prompts is None)
                    first_open = False
                out = self._chain_step(hidden, pos, prompts=prompts)
                for s in self.spans:
                    self.manager.on_request_success(s.span.peer_id)
                break
            except (RpcError, MissingBlocksError, *_RETRYABLE) as e:
                attempt += 1
                self._ban_dead_spans()  # probe BEFORE dropping the spans
                for s in self.spans:
                    s.close()
                self.spans = []
                max_r = self.config.max_retries
                if max_r is not None and attempt > max_r:
                    raise
                delay = self.manager.get_retry_delay(attempt)
                logger.warning("chain step failed (%s); rebuilding route in "
                               "%.1fs (attempt %d)", e, delay, attempt)
                time.sleep(delay)
                self.manager.update()
                # reopen happens at the top of the loop (also under retry)
        if self.config.keep_history:
            self.history.append((pos, hidden, prompts))
        self.position = pos + hidden.shape[1]
        self.step_count += 1
        return out

    # -- speculative decoding --------------------------------------------
    def spec_step(self, hidden: torch.Tensor, position_ids: torch.Tensor,
                  tree_mask: torch.Tensor, tree: Optional[dict] = None):
        """Tree-verify step: hidden (B, T, H) linearized tree nodes with
        per-node absolute positions + ancestor mask. KV written speculatively
        on every span; follow with spec_commit(keep). Spec sessions are
        opened with allow_push=False (the tree tensors ride the client
        stream span by span)."""
        if self.batch_size is None:
            self.batch_size = hidden.shape[0]
        t = self.config.step_timeout
        attempt = 0
        while True:
            try:
                if not self.spans:
                    # spec rides the client stream span by span
                    self.allow_push = False
                    self._open_chain(self.batch_size,
                                     replay=bool(self.history))
                out = hidden
                keep = None
                from bloombee_amd.utils.fault_injection import maybe_fail
                for s in self.spans:
                    maybe_fail("spec_item")

                    async def go(s=s, out=out):
                        await s.stream.send(
                            {"pos": self.position, "spec": True,
                             "step": self.step_count, "tree": tree},
                            [out, position_ids.int(), tree_mask])
                        item = await s.stream.recv()
                        if item is None:
                            raise RpcError(f"stream closed by {s.span.peer_id}")
                        return item[0].get("keep"), item[1][0]

                    keep, out = run_coroutine(go(), t)
                # kept for spec_commit's history entry (failover replay)
                self._last_spec_hidden = hidden
                self.step_count += 1
                return out, keep
            except (RpcError, MissingBlocksError, *_RETRYABLE) as e:
                # rebuilt sessions hold only committed tokens (replayed from
                # history), so the WHOLE tree step can simply be redone
                attempt += 1
                self._ban_dead_spans()
                for s in self.spans:
                    s.close()
                self.spans = []
                max_r = self.config.max_retries
                if max_r is not None and attempt > max_r:
                    raise
                delay = self.manager.get_retry_delay(attempt)
                logger.warning("spec step failed (%s); rebuilding route in "
                               "%.1fs", e, delay)
                time.sleep(delay)
                self.manager.update()

    def spec_commit(self, keep) -> int:
        """Accept tree nodes `keep[b]` (ascending linear indices); every span
        compacts + commits its KV. Returns the accepted length.

        The accepted rows' INPUT hiddens enter the session history first, so
        a failover (here, or in any later step) replays them as ordinary
        committed tokens — positions are sequential after compaction, which
        is exactly the replayed-step semantics."""
        lasth = getattr(self, "_last_spec_hidden", None)
        if lasth is not None and keep and len(keep[0]) > 0:
            kept = lasth[:, list(keep[0])]
            self.history.append((self.position, kept, None))
        t = self.config.step_timeout
        remaining = list(self.spans)
        while remaining:
            s = remaining[0]

            async def go(s=s):
                await s.stream.send({"spec_commit": [list(k) for k in keep]})
                item = await s.stream.recv()
                if item is None or not item[0].get("committed"):
                    raise RpcError(f"spec commit failed on {s.span.peer_id}")

            try:
                run_coroutine(go(), t)
                remaining.pop(0)
            except (RpcError, MissingBlocksError, *_RETRYABLE) as e:
                # rebuild: the replay (which now includes the kept rows)
                # leaves EVERY new span consistent — nothing left to commit
                logger.warning("spec commit failed (%s); rebuilding chain", e)
                self._ban_dead_spans()
                for sp in self.spans:
                    sp.close()
                self.spans = []
                self.manager.update()
                self._open_chain(self.batch_size, replay=True)
                remaining = []
        n = len(keep[0])
        self.position += n
        return n

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()

    def close(self):
        if not self._closed:
            for s in self.spans:
                s.close()
            self._closed = True
