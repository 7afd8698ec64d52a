"""RPC surface of a worker: the reference TransformerConnectionHandler
(server/handler.py:798+ rpc_inference, :2860-3010 rpc_forward/backward,
:1850 rpc_push, :3256 rpc_info) on bloombee_amd.net.rpc.

One asyncio process serves every connection; compute hops to the backend's
worker thread via its prioritized pool, so handler concurrency never blocks
on the GPU (the reference needed 8 forked handler processes + an mp event
bus for this — SURVEY.md §7 step 3 collapses that).

Session ownership: a decode session is created by the first `rpc_inference`
stream item (or an upstream `rpc_push`) and keyed by the client-chosen
session_id, so server→server pushed inputs and client stream inputs meet in
one place (`_iterate_inference_steps` merge semantics, handler.py:1677-1847).
"""
from __future__ import annotations

import asyncio
import uuid
from typing import Dict, List, Optional, Tuple

import torch

from bloombee_amd.net.rpc import RpcClient, RpcServer, Stream
from bloombee_amd.server.backend import StackBackend
from bloombee_amd.utils.logging import get_logger
from bloombee_amd.utils.telemetry import StageTimes

logger = get_logger(__name__)


class AdaptivePushConcurrency:
    """Self-tuning in-flight limit for server->server pushes (parity:
    reference handler.py:255-370): grows the window while pushes succeed
    quickly, shrinks it multiplicatively on failures/slowdowns."""

    def __init__(self, initial: int = 2, lo: int = 1, hi: int = 16,
                 slow_s: float = 0.25):
        self.limit = initial
        self.lo, self.hi = lo, hi
        self.slow_s = slow_s
        self._sem = asyncio.Semaphore(initial)

    async def __aenter__(self):
        await self._sem.acquire()
        return self

    async def __aexit__(self, *exc):
        self._sem.release()

    def feedback(self, elapsed_s: float, ok: bool):
        old = self.limit
        if not ok or elapsed_s > self.slow_s:
            self.limit = max(self.lo, self.limit // 2)
        elif elapsed_s < self.slow_s / 4:
            self.limit = min(self.hi, self.limit + 1)
        for _ in range(self.limit - old):
            self._sem.release()
        for _ in range(old - self.limit):
            # shrink lazily: steal permits as they come back
            asyncio.ensure_future(self._sem.acquire())


class ConnectionHandler:
    def __init__(self, backend: StackBackend, server: RpcServer):
        # micro-batch policy from the typed config (BBAMD_MICROBATCH /
        # _MICRO_BATCH_SIZE / _MIN_BATCH_TO_SPLIT — ref microbatch_config.py)
        from bloombee_amd.config import get_config
        mb = get_config().microbatch
        self.MICROBATCH_ENABLED = mb.enabled
        self.MICROBATCH_MIN_BATCH = mb.min_batch_to_split
        self.MICROBATCH_SIZE = mb.micro_batch_size
        self.KV_MULTIPLEX = mb.kv_multiplex
        self.backend = backend
        self.rpc = server
        # (session_id, step) -> queued pushed inputs awaiting the local stream
        self._push_q: Dict[str, asyncio.Queue] = {}
        self._peers: Dict[Tuple[str, int], RpcClient] = {}
        self._push_limiter = AdaptivePushConcurrency()
        server.register("rpc_info", self.rpc_info)
        server.register("rpc_forward", self.rpc_forward)
        server.register("rpc_backward", self.rpc_backward)
        server.register("rpc_push", self.rpc_push)
        server.register_stream("rpc_inference", self.rpc_inference)
        server.register_stream("rpc_forward_stream", self.rpc_forward_stream)
        server.register_stream("rpc_backward_stream", self.rpc_backward_stream)

    # ------------------------------------------------------------------
    async def rpc_info(self, meta, tensors):
        from bloombee_amd.net.channels import channels
        info = self.backend.info()
        info["dist_rank"] = channels.rank
        return info, []

    @staticmethod
    def _reply_move(meta):
        """Keep reply tensors on-device when they ride the device plane."""
        from bloombee_amd.net.channels import channels
        if channels.enabled and meta.get("drank") is not None:
            return lambda t: t
        return lambda t: t.cpu()

    async def rpc_forward(self, meta, tensors):
        # tensors: [hidden] or [hidden, deep_prompts(n_local_blocks, pre, H)]
        hidden = tensors[0]
        prompts = tensors[1] if len(tensors) > 1 else None
        adapter = meta.get("adapter")
        out = await asyncio.get_event_loop().run_in_executor(
            None, lambda: self.backend.forward(hidden, prompts,
                                               adapter=adapter))
        return {}, [self._reply_move(meta)(out)]

    async def rpc_backward(self, meta, tensors):
        # tensors: [hidden_in, grad_out] or [..., deep_prompts]; replies
        # [grad_in] or [grad_in, grad_prompts]
        hidden_in, grad_out = tensors[0], tensors[1]
        prompts = tensors[2] if len(tensors) > 2 else None
        adapter = meta.get("adapter")
        grad_in, grad_p = await asyncio.get_event_loop().run_in_executor(
            None, lambda: self.backend.backward(hidden_in, grad_out, prompts,
                                                adapter=adapter))
        move = self._reply_move(meta)
        outs = [move(grad_in)]
        if grad_p is not None:
            outs.append(move(grad_p))
        return {}, outs

    # ------------------------------------------------------------------
    async def rpc_forward_stream(self, meta, tensors, stream):
        """Chunked-payload forward (ref rpc_forward_stream, handler.py:
        2860-3010): inputs arrive as flattened parts, the reply streams
        back the same way."""
        from bloombee_amd.net.streaming import (recv_tensors_chunked,
                                                send_tensors_chunked)
        _, ins = await recv_tensors_chunked(stream)
        out_meta, outs = await self.rpc_forward(meta, ins)
        await send_tensors_chunked(stream, outs, meta=out_meta)
        await stream.send_end()

    async def rpc_backward_stream(self, meta, tensors, stream):
        from bloombee_amd.net.streaming import (recv_tensors_chunked,
                                                send_tensors_chunked)
        _, ins = await recv_tensors_chunked(stream)
        out_meta, outs = await self.rpc_backward(meta, ins)
        await send_tensors_chunked(stream, outs, meta=out_meta)
        await stream.send_end()

    # ------------------------------------------------------------------
    async def rpc_push(self, meta, tensors):
        """Upstream server pushed this step's input for a session we serve
        (ref handler.py:1850-1911). Enqueue; the session's inference loop
        prefers pushed inputs over the client stream."""
        sid = meta["session_id"]
        q = self._push_q.setdefault(sid, asyncio.Queue())
        q.put_nowait((meta, tensors))
        return {"ok": True}, []

    # ------------------------------------------------------------------
    async def rpc_inference(self, meta, tensors, stream: Stream):
        """Bidirectional decode stream.

        Open meta: {session_id?, max_length, batch_size, adapter?, codec?,
        push_to?: [host, port, session_id], push_only_recv?, quiet?}.

        Item wire format (meta + tensors):
          * plain step:  {pos, step} + [hidden]           (+ optional deep
            p-tune prompts tensor appended: (n_local_blocks, pre, H))
          * spec step:   {pos, step, spec: true, tree?} +
                         [hidden, position_ids, tree_mask]
          * commit:      {spec_commit: keep}              (no tensors)
          * close:       {close: true}
          * pushed micro-batch (via rpc_push): adds mb: {offset, total}
        Replies: {pos, step, keep?} + [hidden_out]; quiet middle spans in
        push mode reply nothing (the LAST span answers the client)."""
        sid = meta.get("session_id") or uuid.uuid4().hex
        adapter = meta.get("adapter")
        codec = meta.get("codec", "raw")
        max_length = int(meta["max_length"])
        batch_size = int(meta["batch_size"])
        # downstream [host, port, session_id, dist_rank?]
        push_to = meta.get("push_to")
        push_only_recv = bool(meta.get("push_only_recv"))  # inputs via rpc_push
        quiet = bool(meta.get("quiet"))  # don't echo outputs to the client
        # device data plane: reply to the client's dist rank / push to the
        # downstream's dist rank over RCCL when a shared world exists
        from bloombee_amd.net.channels import channels
        if channels.enabled:
            stream.dist_rank = meta.get("drank")
        push_rank = (push_to[3] if push_to and len(push_to) > 3
                     and channels.enabled else None)
        # outputs stay on-device when every consumer rides the device plane;
        # otherwise hop to CPU once and serialize from there
        keep_dev = (stream.dist_rank is not None or quiet) and (
            push_to is None or push_rank is not None)
        move = (lambda t: t) if keep_dev else (lambda t: t.cpu())
        loop = asyncio.get_event_loop()
        times = StageTimes()
        mb_buffers: Dict[int, dict] = {}  # pos -> {offset: out_cpu}
        mb_stage: Dict[int, object] = {}  # offset -> pending swap-out future
        # KV multiplexing: only ~2 micro-batch row windows stay device-
        # resident; admission reserves just that window
        mbs = self.MICROBATCH_SIZE
        multiplex = (self.KV_MULTIPLEX and self.MICROBATCH_ENABLED
                     and batch_size >= self.MICROBATCH_MIN_BATCH
                     and batch_size % mbs == 0 and batch_size > 2 * mbs)
        resident_batch = 3 * mbs if multiplex else None  # j-1 out, j, j+1 in
        # persistent reader futures for push_only_recv: cancelling a q.get()
        # that already dequeued an item silently drops that frame (lost step
        # or spec_commit -> stalled session). Instead the
        # losing future stays pending and is reused next iteration.
        get_push: Optional[asyncio.Future] = None
        get_cli: Optional[asyncio.Future] = None
        await loop.run_in_executor(
            None, lambda: self.backend.open_session(
                sid, batch_size, max_length,
                resident_batch=resident_batch))
        try:
            await stream.send({"session_id": sid, "ok": True})
            while True:
                if push_only_recv:
                    q = self._push_q.setdefault(sid, asyncio.Queue())
                    if get_push is None:
                        get_push = asyncio.ensure_future(q.get())
                    if get_cli is None:
                        get_cli = asyncio.ensure_future(stream.recv())
                    await asyncio.wait({get_push, get_cli},
                                       return_when=asyncio.FIRST_COMPLETED)
                    if get_push.done():  # push preferred (ref merge order)
                        item_meta, item_tensors = get_push.result()
                        get_push = None
                    else:
                        item = get_cli.result()
                        get_cli = None
                        if item is None:
                            break
                        item_meta, item_tensors = item
                        if item_meta.get("close"):
                            break
                else:
                    item = await stream.recv()
                    if item is None:
                        break
                    item_meta, item_tensors = item
                    if item_meta.get("close"):
                        break
                if item_meta.get("spec_commit") is not None:
                    keep = item_meta["spec_commit"]
                    await loop.run_in_executor(
                        None, lambda: self.backend.spec_commit(sid, keep))
                    await stream.send({"committed": True})
                    continue
                pos = int(item_meta["pos"])
                hidden = item_tensors[0]
                spec = bool(item_meta.get("spec"))
                position_ids = tree_mask = prompts = None
                rest = item_tensors[1:]
                if spec:
                    position_ids, tree_mask = rest[0], rest[1]
                    rest = rest[2:]
                if rest:
                    prompts = rest[0]
                def _step():
                    return self.backend.inference_step(
                        sid, hidden, pos, prompts, position_ids, tree_mask,
                        speculative=spec, adapter=adapter)

                mbinfo = item_meta.get("mb")
                if mbinfo is not None and not spec:
                    # pushed micro-batch slice: compute immediately (cross-
                    # stage overlap, ref handler.py:1677-1847 push-preferred
                    # merge), forward the slice downstream, reply merged once
                    # every slice of this step has arrived
                    off = int(mbinfo["offset"])
                    if multiplex:
                        # this stage's KV staging for pushed slices: swap
                        # the slice's row window in before compute (after
                        # any still-running swap-out of the same rows), out
                        # after, prefetching the next expected window
                        h_kv = self.backend.session_handle(sid)
                        nrows = hidden.shape[0]
                        total = int(mbinfo["total"])

                        async def _stage_in(a, n):
                            prev = mb_stage.pop(a, None)
                            if prev is not None:
                                await prev
                            await loop.run_in_executor(
                                None, h_kv.swap_in_rows, a, a + n)

                        await _stage_in(off, nrows)
                        nxt = off + nrows
                        if nxt < total:
                            asyncio.ensure_future(
                                _stage_in(nxt, min(nrows, total - nxt)))
                    part = await loop.run_in_executor(
                        None, lambda: self.backend.inference_step(
                            sid, hidden, pos, prompts, batch_offset=off,
                            adapter=adapter))
                    if multiplex:
                        mb_stage[off] = loop.run_in_executor(
                            None, h_kv.swap_out_rows, off,
                            off + hidden.shape[0])
                    part_cpu = move(part)
                    if push_to is not None:
                        asyncio.ensure_future(self._push_downstream(
                            push_to, pos, part_cpu, item_meta, mb=mbinfo,
                            codec=codec, dist_rank=push_rank))
                    buf = mb_buffers.setdefault(pos, {})
                    buf[off] = part_cpu
                    got = sum(t.shape[0] for t in buf.values())
                    times.bump_step()
                    if got >= int(mbinfo["total"]):
                        out_cpu = torch.cat(
                            [buf[o] for o in sorted(buf)], dim=0)
                        del mb_buffers[pos]
                        if not quiet:
                            await stream.send(
                                {"pos": pos, "step": item_meta.get("step"),
                                 "keep": None}, [out_cpu], codec=codec)
                    continue
                B = hidden.shape[0]
                can_split = (self.MICROBATCH_ENABLED and not spec
                             and push_to is not None
                             and B >= self.MICROBATCH_MIN_BATCH
                             and B % self.MICROBATCH_SIZE == 0)
                if can_split:
                    from bloombee_amd.utils.logging import debug_log
                    debug_log("microbatch", logger,
                              "splitting B=%d into %d-row micro-batches", B,
                              self.MICROBATCH_SIZE)
                    # reference micro-batch overlap (block_functions.py:
                    # 2055-2460): compute micro-batch j while j-1 is in
                    # flight downstream; the client reply carries the merged
                    # batch from the LAST span as usual
                    mbs = self.MICROBATCH_SIZE
                    outs = []
                    push_tasks = []
                    stage_tasks = []
                    h_kv = (self.backend.session_handle(sid) if multiplex
                            else None)

                    def _swap_in(a):
                        h_kv.swap_in_rows(a, min(a + mbs, B))

                    def _swap_out(a):
                        h_kv.swap_out_rows(a, min(a + mbs, B))

                    with times.span("compute"):
                        # KV staging pipeline (ref memory_cache_manager.py:
                        # 944-1371): while slice j computes, slice j+1's KV
                        # swaps in on the staging stream and slice j-1's
                        # swaps out — only the resident window holds pages
                        pf = (loop.run_in_executor(None, _swap_in, 0)
                              if multiplex else None)
                        for j in range(0, B, mbs):
                            if pf is not None:
                                await pf  # slice j resident
                            if multiplex and j + mbs < B:
                                pf = loop.run_in_executor(
                                    None, _swap_in, j + mbs)
                            else:
                                pf = None
                            part = await loop.run_in_executor(
                                None, lambda j=j: self.backend.inference_step(
                                    sid, hidden[j:j + mbs], pos, prompts,
                                    batch_offset=j, adapter=adapter))
                            part_cpu = move(part)
                            if multiplex:
                                stage_tasks.append(loop.run_in_executor(
                                    None, _swap_out, j))
                            outs.append(part_cpu)
                            push_tasks.append(asyncio.ensure_future(
                                self._push_downstream(
                                    push_to, pos, part_cpu, item_meta,
                                    mb={"offset": j, "total": B},
                                    codec=codec, dist_rank=push_rank)))
                        out_cpu = torch.cat(outs, dim=0)
                        for t_ in push_tasks:
                            await t_
                        for t_ in stage_tasks:
                            await t_
                    times.bump_step()
                    if not quiet:
                        with times.span("reply"):
                            await stream.send(
                                {"pos": pos, "step": item_meta.get("step"),
                                 "keep": None}, [out_cpu], codec=codec)
                    continue
                with times.span("compute"):
                    out = await loop.run_in_executor(None, _step)
                keep = None
                tree = item_meta.get("tree")
                if (spec and tree is not None
                        and self.backend.pruner is not None):
                    out, keep = await loop.run_in_executor(
                        None, lambda: self.backend.prune_tree(
                            out, tree["tokens"], tree["parents"]))
                out_cpu = move(out)
                times.bump_step()
                if push_to is not None:
                    with times.span("push"):
                        await self._push_downstream(push_to, pos, out_cpu,
                                                    item_meta, codec=codec,
                                                    dist_rank=push_rank)
                if not quiet:
                    with times.span("reply"):
                        await stream.send(
                            {"pos": pos, "step": item_meta.get("step"),
                             "keep": keep},
                            [out_cpu], codec=codec)
                # quiet spans send nothing: during push-mode decode the client
                # reads outputs from the LAST span only (push_only_downstream,
                # ref inference_session.py:178-196)
        finally:
            for fut in (get_push, get_cli):
                if fut is not None:
                    fut.cancel()
            if times.steps:
                logger.info("session %s closed\n%s", sid[:8], times.table())
            await loop.run_in_executor(None,
                                       lambda: self.backend.close_session(sid))
            self._push_q.pop(sid, None)
            try:
                await stream.send_end()
            except Exception:
                pass

    async def _push_downstream(self, push_to, pos: int, hidden: torch.Tensor,
                               item_meta: dict, mb: Optional[dict] = None,
                               codec: str = "raw",
                               dist_rank: Optional[int] = None) -> None:
        import time as _time

        from bloombee_amd.utils.fault_injection import maybe_fail
        try:
            maybe_fail("s2s_push")
        except Exception as e:
            logger.warning("s2s push suppressed by fault injection: %s", e)
            return
        host, port, down_sid = push_to[0], int(push_to[1]), push_to[2]
        key = (host, port)
        if key not in self._peers:
            self._peers[key] = RpcClient(host, port)
        t0 = _time.monotonic()
        ok = True
        try:
            async with self._push_limiter:
                await self._peers[key].call(
                    "rpc_push", {"session_id": down_sid, "pos": pos,
                                 "step": item_meta.get("step"),
                                 "mb": mb}, [hidden],
                    codec=codec, timeout=30, dist_rank=dist_rank)
        except Exception as e:  # noqa: BLE001 — client will fall back
            ok = False
            logger.warning("s2s push to %s failed: %s", push_to, e)
        self._push_limiter.feedback(_time.monotonic() - t0, ok)
