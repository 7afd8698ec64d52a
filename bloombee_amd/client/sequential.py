"""RemoteSequential: an nn.Module whose forward runs on the swarm.

Parity: reference client/remote_sequential.py + sequential_autograd.py —
fault-tolerant span-wise forward with retry/re-routing, and a backward that
re-runs forward on replacement servers to rebuild lost activations
(sequential_autograd.py:96-152).
"""
from __future__ import annotations

import time
from typing import List, Optional, Tuple

import torch

from bloombee_amd.client.config import ClientConfig
from bloombee_amd.client.routing import (MissingBlocksError,
                                         RemoteSequenceManager,
                                         RemoteSpanInfo)
from bloombee_amd.client.session import InferenceSession
from bloombee_amd.client.worker import get_client, run_coroutine
from bloombee_amd.net.rpc import RpcError
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)

# On Python < 3.11, concurrent.futures.TimeoutError and asyncio.TimeoutError
# are NOT the builtin TimeoutError — a step timeout must still trigger
# failover (found by fault injection on the s2s push path)
import asyncio as _asyncio  # noqa: E402
import concurrent.futures as _cf  # noqa: E402

_RETRYABLE = (OSError, TimeoutError, ConnectionError,
              _cf.TimeoutError, _asyncio.TimeoutError)

MAX_TOKENS_IN_BATCH = 1024  # client sub-batch split (ref sequential_autograd.py:22)


def _use_stream(span: RemoteSpanInfo, payload) -> bool:
    """Oversized payloads stream in chunks (ref remote_forward_backward.py:
    115-118 unary-vs-stream split) — unless the peer shares our device
    plane, where the payload rides RCCL out-of-band anyway."""
    from bloombee_amd.net.channels import channels
    from bloombee_amd.net.streaming import (MAX_UNARY_PAYLOAD_BYTES,
                                            payload_nbytes)
    if channels.enabled and span.server_info.dist_rank is not None:
        return False
    return payload_nbytes(payload) > MAX_UNARY_PAYLOAD_BYTES


def _call_streamed(span: RemoteSpanInfo, method: str, payload, timeout: float,
                   adapter: Optional[str]):
    from bloombee_amd.net.streaming import (recv_tensors_chunked,
                                            send_tensors_chunked)
    client = get_client(span.server_info.host, span.server_info.port)

    async def go():
        stream = await client.open_stream(method, {"adapter": adapter})
        await send_tensors_chunked(stream, payload)
        meta, tensors = await recv_tensors_chunked(stream)
        return tensors

    return run_coroutine(go(), timeout + 5)


def _call_forward(span: RemoteSpanInfo, hidden: torch.Tensor,
                  timeout: float,
                  prompts: Optional[torch.Tensor] = None,
                  adapter: Optional[str] = None) -> torch.Tensor:
    from bloombee_amd.net.channels import channels
    client = get_client(span.server_info.host, span.server_info.port)
    payload = [hidden] if prompts is None else [hidden, prompts]
    if _use_stream(span, payload):
        return _call_streamed(span, "rpc_forward_stream", payload, timeout,
                              adapter)[0]
    dist_rank = (span.server_info.dist_rank if channels.enabled else None)
    meta, tensors = run_coroutine(
        client.call("rpc_forward", {"adapter": adapter,
                                    "drank": channels.rank}, payload,
                    timeout=timeout, dist_rank=dist_rank),
        timeout + 5)
    return tensors[0]


def _call_backward(span: RemoteSpanInfo, hidden_in: torch.Tensor,
                   grad_out: torch.Tensor, timeout: float,
                   prompts: Optional[torch.Tensor] = None,
                   adapter: Optional[str] = None,
                   ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    from bloombee_amd.net.channels import channels
    client = get_client(span.server_info.host, span.server_info.port)
    payload = ([hidden_in, grad_out] if prompts is None
               else [hidden_in, grad_out, prompts])
    if _use_stream(span, payload):
        tensors = _call_streamed(span, "rpc_backward_stream", payload,
                                 timeout, adapter)
        return tensors[0], (tensors[1] if len(tensors) > 1 else None)
    dist_rank = (span.server_info.dist_rank if channels.enabled else None)
    meta, tensors = run_coroutine(
        client.call("rpc_backward", {"adapter": adapter,
                                     "drank": channels.rank}, payload,
                    timeout=timeout, dist_rank=dist_rank),
        timeout + 5)
    return tensors[0], (tensors[1] if len(tensors) > 1 else None)


def sequential_forward(manager: RemoteSequenceManager, hidden: torch.Tensor,
                       start: int = 0, end: Optional[int] = None,
                       prompts: Optional[torch.Tensor] = None,
                       ) -> Tuple[torch.Tensor, List[Tuple[RemoteSpanInfo, torch.Tensor]]]:
    """Forward through [start, end); returns output + per-span (span, input)
    pairs saved for backward (ref sequential_forward :25-105). prompts:
    full-depth deep p-tune tensor (num_blocks, pre, H); each span receives
    its block slice."""
    end = end if end is not None else manager.num_blocks
    cfg = manager.config
    saved: List[Tuple[RemoteSpanInfo, torch.Tensor]] = []
    cur = start
    out = hidden
    attempt = 0
    while cur < end:
        span = None
        try:
            # transient MissingBlocks (e.g. every server briefly banned)
            # must wait out the ban backoff, not abort the pass
            route = manager.make_sequence(cur, end)
            span = route[0]
            p = (prompts[span.start:span.end].detach()
                 if prompts is not None else None)
            result = _call_forward(span, out, cfg.request_timeout, prompts=p,
                                   adapter=getattr(cfg, "active_adapter",
                                                   None))
            manager.on_request_success(span.peer_id)
            saved.append((span, out))
            out = result
            cur = span.end
            attempt = 0
        except (RpcError, MissingBlocksError, *_RETRYABLE) as e:
            attempt += 1
            if span is not None:
                manager.on_request_failure(span.peer_id)
            if cfg.max_retries is not None and attempt > cfg.max_retries:
                raise
            delay = manager.get_retry_delay(attempt)
            logger.warning("forward pass at block %d failed (%s); retrying "
                           "in %.1fs", cur, e, delay)
            time.sleep(delay)
            manager.update()
    return out, saved


def sequential_backward(manager: RemoteSequenceManager, grad_out: torch.Tensor,
                        saved: List[Tuple[RemoteSpanInfo, torch.Tensor]],
                        prompts: Optional[torch.Tensor] = None,
                        grad_prompts: Optional[torch.Tensor] = None,
                        ) -> torch.Tensor:
    """Reverse pass over the saved spans; on failure re-routes the span and
    re-runs forward from its saved input to rebuild activations on the
    replacement server (ref sequential_backward :112-197). When prompts is
    given, per-span deep-prompt grads are accumulated into grad_prompts
    (full-depth, pre-zeroed) in place."""
    cfg = manager.config
    grad = grad_out
    for span, span_input in reversed(saved):
        attempt = 0
        while True:
            try:
                p = (prompts[span.start:span.end].detach()
                     if prompts is not None else None)
                grad, gp = _call_backward(span, span_input, grad,
                                          cfg.request_timeout, prompts=p,
                                          adapter=getattr(cfg,
                                                          "active_adapter",
                                                          None))
                if gp is not None and grad_prompts is not None:
                    grad_prompts[span.start:span.end] += gp.to(grad_prompts.dtype)
                manager.on_request_success(span.peer_id)
                break
            except (RpcError, MissingBlocksError, *_RETRYABLE) as e:
                attempt += 1
                manager.on_request_failure(span.peer_id)
                if cfg.max_retries is not None and attempt > cfg.max_retries:
                    raise
                time.sleep(manager.get_retry_delay(attempt))
                manager.update()
                # find a replacement covering exactly this span's range; the
                # replacement recomputes the forward internally in
                # rpc_backward. A still-banned route raises MissingBlocks
                # again -> caught above on the next spin.
                try:
                    route = manager.make_sequence(span.start, span.end)
                except MissingBlocksError:
                    continue
                if len(route) == 1:
                    span = route[0]
                else:
                    # range now split across servers: forward to rebuild
                    # intermediate inputs, then continue span-wise
                    logger.warning("backward span re-split across %d servers",
                                   len(route))
                    _, sub_saved = sequential_forward(
                        manager, span_input, span.start, span.end,
                        prompts=prompts)
                    grad = sequential_backward(manager, grad, sub_saved,
                                               prompts=prompts,
                                               grad_prompts=grad_prompts)
                    span = None
                    break
        if span is None:
            continue
    return grad


class _RemoteSequentialAutograd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, hidden: torch.Tensor,
                prompts: Optional[torch.Tensor],
                manager: RemoteSequenceManager):
        outs, all_saved = [], []
        # token-bounded sub-batches (ref MAX_TOKENS_IN_BATCH)
        B, T, H = hidden.shape
        rows = max(1, MAX_TOKENS_IN_BATCH // max(T, 1))
        for i in range(0, B, rows):
            out, saved = sequential_forward(manager, hidden[i:i + rows].detach(),
                                            prompts=prompts)
            outs.append(out)
            all_saved.append(saved)
        ctx.manager = manager
        ctx.saved = all_saved
        ctx.rows = rows
        ctx.prompts = prompts
        return torch.cat(outs, dim=0)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        grads = []
        gp = (torch.zeros_like(ctx.prompts, dtype=torch.float32)
              if ctx.prompts is not None else None)
        for i, saved in enumerate(ctx.saved):
            g = grad_out[i * ctx.rows:(i + 1) * ctx.rows]
            grads.append(sequential_backward(ctx.manager, g, saved,
                                             prompts=ctx.prompts,
                                             grad_prompts=gp))
        gp_out = gp.to(ctx.prompts.dtype) if gp is not None else None
        return torch.cat(grads, dim=0), gp_out, None


class RemoteSequential(torch.nn.Module):
    """Chain of remote transformer blocks (ref client/remote_sequential.py)."""

    def __init__(self, config: ClientConfig, model_name: str, num_blocks: int,
                 manager: Optional[RemoteSequenceManager] = None):
        super().__init__()
        self.config = config
        self.manager = manager or RemoteSequenceManager(config, model_name,
                                                        num_blocks)
        self._active_session: Optional[InferenceSession] = None

    def forward(self, hidden: torch.Tensor,
                prompts: Optional[torch.Tensor] = None) -> torch.Tensor:
        if self._active_session is not None:
            return self._active_session.step(hidden)
        return _RemoteSequentialAutograd.apply(hidden, prompts, self.manager)

    def inference_session(self, max_length: int) -> InferenceSession:
        return InferenceSession(self.manager, max_length)

    class _Use:
        def __init__(self, outer, session):
            self.outer, self.session = outer, session

        def __enter__(self):
            self.outer._active_session = self.session
            return self.session

        def __exit__(self, *exc):
            self.outer._active_session = None

    def use_session(self, session: InferenceSession) -> "_Use":
        return RemoteSequential._Use(self, session)
