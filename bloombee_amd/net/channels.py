"""Device-resident tensor channels: the RCCL/xGMI data plane of the swarm.

The reference moves every inter-worker activation GPU→CPU→serialize→NIC→
CPU→GPU and models that hop cost explicitly (reference server/handler.py:
1584-1605 [TIMING_NOTE]). On one 8×MI355X node those hops are exactly what
RCCL send/recv over xGMI eliminates (SURVEY.md §5 "Distributed communication
backend", §7 hard-part 6): when two swarm processes share a torch.distributed
world (one process per GPU, backend "nccl" = RCCL on ROCm), tensor payloads
stay on-device and ride point-to-point xGMI links, while the existing TCP
frames keep carrying the control metadata + a small per-tensor descriptor.
Off-node peers (no shared world) keep the TCP payload path untouched — one
channel abstraction, no dual code paths above it.

Ordering contract: RCCL point-to-point ops between a (src, dst) pair must be
posted in the same order on both sides (NCCL ignores tags). The sender
assigns a per-destination sequence number under a lock and the pump thread
posts isends in assignment order; the receiver's pump holds out-of-order
recv requests and posts irecvs strictly in sequence order per source. This
makes the data plane correct even when several TCP connections (client
stream, server→server push) interleave between the same process pair.

Completion model:
  * NCCL/RCCL: Work.wait() only inserts a stream dependency (non-blocking on
    the CPU), so a recv future resolves as soon as the irecv is posted — any
    later kernel on the device's default stream is ordered after the data
    lands. Sent tensors are kept alive until a post-op HIP event clears.
  * gloo (CPU tests): gloo p2p works never self-report completion, so each
    posted op is parked on a waiter thread that blocks in Work.wait().
  * same process (dst == own rank — the 1-GPU swarm bench, in-process test
    swarms): the tensor is handed over through a mailbox, zero-copy.
"""
from __future__ import annotations

import collections
import threading
import time
from concurrent.futures import Future
from typing import Deque, Dict, List, Optional, Tuple

import torch

from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


class _Req:
    __slots__ = ("kind", "tensor", "peer", "seq", "fut")

    def __init__(self, kind: str, tensor: torch.Tensor, peer: int, seq: int,
                 fut: Optional[Future] = None):
        self.kind = kind          # "send" | "recv"
        self.tensor = tensor      # payload (send) or target buffer (recv)
        self.peer = peer          # dst (send) / src (recv) rank
        self.seq = seq
        self.fut = fut


class DistChannels:
    """Process-wide singleton (module-level `channels`)."""

    def __init__(self):
        self._device: Optional[torch.device] = None
        self._rank: Optional[int] = None
        self._world: int = 1
        self._on = False
        self._use_dist = False           # False => world of 1, mailbox only
        self._cv = threading.Condition()
        self._send_seq: Dict[int, int] = {}
        self._queue: Deque[_Req] = collections.deque()
        self._self_box: Dict[int, torch.Tensor] = {}
        self._self_futs: Dict[int, Future] = {}
        self._recv_next: Dict[int, int] = {}
        self._recv_hold: Dict[int, Dict[int, _Req]] = {}
        self._gc: Deque[Tuple[object, torch.Tensor]] = collections.deque()
        self._waiters = None  # CPU-path completion threads (see _post)
        self._thread: Optional[threading.Thread] = None
        self._group = None  # dedicated process group (see enable)

    # -- lifecycle --------------------------------------------------------
    def enable(self, device) -> None:
        """Turn the device data plane on. If torch.distributed is initialized
        a DEDICATED process group is created for the channel (collective call
        — every rank must enable); otherwise a standalone world of one is
        assumed (rank 0, mailbox delivery only — the 1-GPU swarm where client
        and workers share the process).

        The dedicated group keeps pump-thread p2p ops off the default group:
        on NCCL/RCCL, interleaving them with user collectives (bench
        barriers, TP all-reduce) from other threads can deadlock the
        communicator; on gloo, a blocking barrier would starve the small
        per-group op thread pool the p2p ops also need."""
        import torch.distributed as dist

        group = None
        if dist.is_initialized() and dist.get_world_size() > 1:
            opts = None
            if dist.get_backend() == "gloo":
                # widen the gloo op pool: the pump keeps several p2p ops in
                # flight (replies + pushes + next-step recvs)
                opts = dist.ProcessGroupGloo._Options()
                opts._threads = 16
            group = dist.new_group(backend=dist.get_backend(),
                                   pg_options=opts)
        with self._cv:
            if self._on:
                return
            self._device = torch.device(device)
            self._use_dist = dist.is_initialized()
            self._rank = dist.get_rank() if self._use_dist else 0
            self._world = dist.get_world_size() if self._use_dist else 1
            self._group = group
            self._on = True
        if self._use_dist and self._world > 1:
            if self._device.type != "cuda":
                from concurrent.futures import ThreadPoolExecutor
                self._waiters = ThreadPoolExecutor(
                    max_workers=32, thread_name_prefix="bbamd.dist-wait")
            self._thread = threading.Thread(target=self._pump, daemon=True,
                                            name="bbamd.dist-pump")
            self._thread.start()
        logger.info("dist channels enabled: rank %d/%d on %s (%s)",
                    self._rank, self._world, self._device,
                    "dist" if self._use_dist else "mailbox")

    def disable(self) -> None:
        with self._cv:
            self._on = False
            self._cv.notify_all()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None
        if self._waiters is not None:
            self._waiters.shutdown(wait=False)
            self._waiters = None
        with self._cv:
            self._send_seq.clear()
            self._self_box.clear()
            self._self_futs.clear()
            self._recv_next.clear()
            self._recv_hold.clear()
            self._queue.clear()
            self._gc.clear()
            self._group = None

    @property
    def enabled(self) -> bool:
        return self._on

    @property
    def rank(self) -> Optional[int]:
        return self._rank if self._on else None

    @property
    def device(self) -> Optional[torch.device]:
        return self._device if self._on else None

    # -- data plane -------------------------------------------------------
    def send(self, t: torch.Tensor, dst: int) -> int:
        """Queue a device send to `dst`; returns the per-destination sequence
        number the receiver must use. Thread-safe; posts are async."""
        if not self._on:
            raise RuntimeError("dist channels not enabled")
        t = t.detach()
        if t.dtype == torch.bool:
            t = t.to(torch.uint8)
        t = t.to(self._device, non_blocking=True).contiguous()
        with self._cv:
            seq = self._send_seq.get(dst, 0)
            self._send_seq[dst] = seq + 1
            if dst == self._rank:
                fut = self._self_futs.pop(seq, None)
                if fut is not None:
                    fut.set_result(t)
                else:
                    self._self_box[seq] = t
                return seq
            self._queue.append(_Req("send", t, dst, seq))
            self._cv.notify()
        return seq

    def recv(self, shape: List[int], dtype: torch.dtype, src: int,
             seq: int) -> Future:
        """Future resolving to the received device tensor. Must be called in
        frame order per connection (the read loops do); cross-connection
        ordering is restored by the per-src sequence numbers."""
        if not self._on:
            raise RuntimeError("dist channels not enabled")
        fut: Future = Future()
        wire_dtype = torch.uint8 if dtype == torch.bool else dtype
        with self._cv:
            if src == self._rank:
                t = self._self_box.pop(seq, None)
                if t is not None:
                    fut.set_result(t)
                else:
                    self._self_futs[seq] = fut
                return fut
            buf = torch.empty(shape, dtype=wire_dtype, device=self._device)
            self._queue.append(_Req("recv", buf, src, seq, fut))
            self._cv.notify()
        return fut

    # -- pump -------------------------------------------------------------
    def _pump(self):
        try:
            self._pump_inner()
        except Exception:  # pragma: no cover - fatal data-plane fault
            logger.exception("dist channel pump died; failing in-flight ops")
            with self._cv:
                reqs = list(self._queue)
                self._queue.clear()
                for hold in self._recv_hold.values():
                    reqs.extend(hold.values())
                self._recv_hold.clear()
            for r in reqs:
                if r.fut is not None and not r.fut.done():
                    r.fut.set_exception(RuntimeError("dist pump died"))

    def _pump_inner(self):
        import torch.distributed as dist

        on_gpu = self._device.type == "cuda"
        if on_gpu:
            torch.cuda.set_device(self._device)
        while True:
            with self._cv:
                while self._on and not self._queue and not self._gc:
                    self._cv.wait(timeout=0.1)
                if not self._on:
                    return
                reqs = list(self._queue)
                self._queue.clear()
            for r in reqs:
                if r.kind == "send":
                    self._post(dist.isend(r.tensor, r.peer,
                                          group=self._group), r, on_gpu)
                else:
                    self._recv_hold.setdefault(r.peer, {})[r.seq] = r
            # post eligible recvs strictly in per-src seq order
            for src, hold in self._recv_hold.items():
                nxt = self._recv_next.get(src, 0)
                while nxt in hold:
                    r = hold.pop(nxt)
                    self._post(dist.irecv(r.tensor, src, group=self._group),
                               r, on_gpu)
                    nxt += 1
                self._recv_next[src] = nxt
            # completed-send GC (GPU): free tensor refs once the event clears
            if on_gpu:
                while self._gc and self._gc[0][0].query():
                    self._gc.popleft()

    def _post(self, work, r: _Req, on_gpu: bool) -> None:
        if on_gpu:
            # RCCL: wait() only makes the current (default) stream wait on
            # the comm stream — non-blocking on the CPU. Consumers launch on
            # the default stream, so the data dependency is already encoded;
            # resolve the future now and keep the tensor alive via an event.
            work.wait()
            ev = torch.cuda.Event()
            ev.record()
            self._gc.append((ev, r.tensor))
            if r.fut is not None and not r.fut.done():
                r.fut.set_result(r.tensor)
        else:
            # gloo Work objects never report is_completed() for p2p without
            # a blocking wait() — park each op on a waiter thread (CPU test
            # path only; bounded by the pool size)
            self._waiters.submit(self._wait_cpu, work, r)

    @staticmethod
    def _wait_cpu(work, r: _Req) -> None:
        try:
            work.wait()
        except Exception as e:  # noqa: BLE001 — surfaced via the future
            if r.fut is not None and not r.fut.done():
                r.fut.set_exception(e)
            return
        if r.fut is not None and not r.fut.done():
            r.fut.set_result(r.tensor)


channels = DistChannels()
