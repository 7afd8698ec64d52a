"""Server compute backend: a BlockStack + paged KV + the prioritized pool.

Parity target: reference TransformerBackend (server/backend.py:488-789
inference_step; :427-462 backward) and _MergedInferenceStep (:1214-1399) —
here the whole local block range always runs as one merged call on the worker
thread, with the paged KV session handle playing the role of the 2,160-line
KVCacheManager (kv/paged.py docstring).
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch

from bloombee_amd.engine import BlockStack
from bloombee_amd.kv.paged import PagedKVCache, SessionHandle
from bloombee_amd.models.base import ModelConfig
from bloombee_amd.server.task_pool import (PRIORITY_INFERENCE, PRIORITY_TRAIN,
                                           TaskPool)
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


@dataclass
class SessionState:
    handle: SessionHandle
    last_activity: float = field(default_factory=time.monotonic)


class StackBackend:
    """One contiguous block range on one device, served concurrently."""

    def __init__(self, config: ModelConfig, start: int, end: int,
                 device: str = "cpu", seed: int = 0,
                 kv_max_tokens: int = 1 << 18,
                 checkpoint_dir: Optional[str] = None,
                 offload_policy=None,
                 max_batch_size: int = 2048):
        self.config = config
        self.start, self.end = start, end
        self.device = torch.device(device)
        self.stack = BlockStack(config, start, end, device=device, seed=seed)
        if checkpoint_dir is not None:
            from bloombee_amd.server.from_pretrained import load_block_weights
            for i, blk in enumerate(self.stack.blocks):
                load_block_weights(blk, checkpoint_dir, start + i)
        if offload_policy is not None and offload_policy.offloads_weights:
            from bloombee_amd.offload.weights import OffloadedBlockStack
            self.stack = OffloadedBlockStack(self.stack, offload_policy)
        if offload_policy is not None and offload_policy.attn_sparsity < 1.0:
            from bloombee_amd import ops as _ops
            _ops.set_attn_sparsity(offload_policy.attn_sparsity)
        self.compress_swap = bool(offload_policy is not None
                                  and offload_policy.compress_cache)
        self.kv_pool: PagedKVCache = self.stack.make_kv(kv_max_tokens)
        self.is_last_block = (end == config.num_hidden_layers)
        self.pruner = None
        if self.is_last_block:
            from bloombee_amd.spec.pruner import (MidLMHead, PruningMethod,
                                                  create_pruner)
            import os as _os

            method = PruningMethod(_os.environ.get("BBAMD_SPEC_PRUNING",
                                                   "none"))
            if method != PruningMethod.NONE:
                head = MidLMHead(config.hidden_size, config.vocab_size,
                                 seed=seed, dtype=config.dtype, device=device)
                self.pruner = create_pruner(method, head)
                logger.info("mid-network tree pruner enabled: %s", method)
        # sequence-chunk bound for long prefills (ref max_chunk_size_bytes)
        import os as _os2
        self.max_chunk_tokens = int(_os2.environ.get("BBAMD_MAX_CHUNK_TOKENS",
                                                     "4096"))
        self.max_batch_size = max_batch_size
        self.pool = TaskPool(name=f"worker[{start}:{end}]")
        self.sessions: Dict[str, SessionState] = {}
        self._lock = threading.Lock()
        # hipGraph decode capture (the server backend
        # step was not captured — only the single-rank pipeline was). The
        # whole local block range replays as one graph per (session, B);
        # dynamic state (token position, page table, KV) lives in device
        # tensors so the same graph serves every decode step.
        from bloombee_amd.config import get_config
        self._use_graphs = (get_config().use_hip_graphs
                            and self.device.type == "cuda"
                            and type(self.stack).__name__ == "BlockStack")
        self._graphs: Dict[str, dict] = {}
        self.graphs_captured = 0  # lifetime count (survives session close)

    # -- sessions ---------------------------------------------------------
    def open_session(self, session_id: str, batch_size: int, max_length: int,
                     timeout: Optional[float] = 10.0,
                     resident_batch: Optional[int] = None) -> None:
        from bloombee_amd.kv.paged import AllocationFailed

        if batch_size > self.max_batch_size:
            raise ValueError(
                f"batch {batch_size} > server max_batch_size "
                f"{self.max_batch_size} (ref --max_batch_size)")

        try:
            handle = self.kv_pool.allocate(batch_size, max_length, timeout=0.5,
                                           resident_batch=resident_batch)
        except AllocationFailed:
            # KV pressure: offload the least-recently-active idle session to
            # host (ref micro-batch KV multiplexing) and retry
            self._swap_out_idle()
            handle = self.kv_pool.allocate(batch_size, max_length,
                                           timeout=timeout,
                                           resident_batch=resident_batch)
        with self._lock:
            self.sessions[session_id] = SessionState(handle)

    def session_handle(self, session_id: str):
        """The session's KV handle — the handler's KV-staging driver uses it
        to prefetch/offload micro-batch row windows."""
        return self._session(session_id).handle

    def _swap_out_idle(self) -> None:
        with self._lock:
            candidates = sorted(
                ((sid, st) for sid, st in self.sessions.items()
                 if not st.handle.is_swapped),
                key=lambda kv: kv[1].last_activity)
        for sid, st in candidates[:1]:
            logger.info("KV pressure: swapping out idle session %s", sid[:8])
            self.pool.submit(
                lambda h=st.handle: h.swap_out(compress=self.compress_swap),
                PRIORITY_TRAIN).result()

    def reap_idle_sessions(self, max_idle_s: float = 600.0) -> int:
        """Close sessions idle past max_idle_s (ref handler session GC —
        a client that vanished without closing its stream must not pin KV
        pages forever). Returns the number reaped."""
        now = time.monotonic()
        with self._lock:
            dead = [sid for sid, st in self.sessions.items()
                    if now - st.last_activity > max_idle_s]
        for sid in dead:
            logger.info("reaping idle session %s", sid[:8])
            self.close_session(sid)
        return len(dead)

    def close_session(self, session_id: str) -> None:
        with self._lock:
            state = self.sessions.pop(session_id, None)
        self._graphs.pop(session_id, None)
        if state is not None:
            state.handle.close()

    def _session(self, session_id: str) -> SessionState:
        with self._lock:
            state = self.sessions.get(session_id)
        if state is None:
            raise KeyError(f"unknown session {session_id!r}")
        state.last_activity = time.monotonic()
        return state

    # -- compute entry points (run on the worker thread) ------------------
    def inference_step(self, session_id: str, hidden: torch.Tensor,
                       start_pos: int, prompts: Optional[torch.Tensor] = None,
                       position_ids: Optional[torch.Tensor] = None,
                       tree_mask: Optional[torch.Tensor] = None,
                       speculative: bool = False,
                       batch_offset: Optional[int] = None,
                       adapter: Optional[str] = None) -> torch.Tensor:
        """One decode/prefill step for an open session.

        start_pos: absolute position of hidden[:, 0]. If the session has
        advanced further (failover replay, ref inference_session.py:802-831),
        the cache is truncated back to start_pos first.

        speculative: the step's tokens are written ABOVE l_acc and stay
        pending until spec_commit (tree verify; ref backend.py:944-1047 tree
        masks + rotary tree positions, paged commit/rollback).
        """
        state = self._session(session_id)
        handle = state.handle

        def run():
            # adapter selection must live on THIS worker thread: the
            # handler's contextvar does not cross the task-pool boundary
            from bloombee_amd.config import get_config
            from bloombee_amd.utils.peft import using_adapter
            from bloombee_amd.utils.trace import trace_range
            with using_adapter(adapter), \
                    trace_range(f"infer[{self.start}:{self.end}] "
                                f"pos={start_pos}"):
                if not get_config().step_profile:
                    return _compute()
                # BBAMD_STEP_PROFILE (ref BLOOMBEE_STEP_PROFILE,
                # backend.py:59-60): wall-clock the step with device sync
                if self.device.type == "cuda":
                    torch.cuda.synchronize(self.device)
                t0 = time.monotonic()
                out_ = _compute()
                if self.device.type == "cuda":
                    torch.cuda.synchronize(self.device)
                logger.info("[STEP_PROFILE] blocks[%d:%d] pos=%d T=%d "
                            "%.3f ms", self.start, self.end, start_pos,
                            hidden.shape[1],
                            (time.monotonic() - t0) * 1e3)
                return out_

        def _compute():
            nonlocal start_pos
            T_ = hidden.shape[1]
            plain_decode = (T_ == 1 and not speculative and tree_mask is None
                            and position_ids is None)
            if handle.is_swapped:
                from bloombee_amd.config import get_config as _gc
                mode = _gc().kv.mixed_attn
                use_mixed = plain_decode and (
                    mode in ("on", True)  # bool tolerated for programmatic cfgs
                    or (mode == "auto" and handle.swapped_pages_needed()
                        > handle.cache.free_page_count()))
                if use_mixed:
                    # capacity mode: committed KV stays host-side; decode
                    # merges it with the device-resident recent segment
                    handle.swap_in_as_prefix()
                else:
                    handle.swap_in()
            # host-prefixed sessions use pool-local positions (rotary stays
            # absolute inside the blocks via kv.pos_offset)
            if handle.pos_offset:
                if hidden.shape[1] > 1 or speculative or tree_mask is not None:
                    raise ValueError(
                        "mixed-device (host-prefix) sessions are decode-only;"
                        " disable BBAMD_MIXED_ATTN for this workload")
                start_pos = start_pos - handle.pos_offset
            h = hidden.to(self.device, non_blocking=True)
            if h.dtype != self.config.dtype:
                h = h.to(self.config.dtype)
            deep = None
            if prompts is not None:
                p = prompts.to(h.device)
                if p.dim() == 3 and p.shape[0] == len(self.stack.blocks):
                    # deep p-tuning: (n_local_blocks, pre, H) added to the
                    # prompt positions at every block input
                    deep = p
                else:
                    # legacy shallow form: additive delta for the span input
                    h = h + p.to(h.dtype)
            B, T, _ = h.shape
            cur = handle.seqs[0].l_acc
            if batch_offset is not None:
                # micro-batch slice of the session (server-side split or
                # upstream per-MB push): each slice extends ONLY its own
                # rows, so a KV-multiplexed session never allocates device
                # pages outside the resident window
                b0, b1 = batch_offset, batch_offset + B
                if handle.seqs[b0].l_spec == start_pos:
                    handle.extend_rows(b0, b1, T)
                elif handle.seqs[b0].l_spec != start_pos + T:
                    raise ValueError(
                        f"micro-batch at position {start_pos} inconsistent "
                        f"with cache length {handle.seqs[b0].l_spec}")
                from bloombee_amd.kv.views import SessionView
                view = SessionView(handle, batch_offset, batch_offset + B)
                sp = torch.full((B,), start_pos, dtype=torch.int32,
                                device=self.device)
                return self.stack.forward_inference(h, view, sp,
                                                    deep_prompts=deep)
            if start_pos < cur:
                handle.truncate([min(start_pos, s.l_acc) for s in handle.seqs])
                handle.rollback()
            elif start_pos > cur:
                raise ValueError(
                    f"inference step at position {start_pos} but cache has "
                    f"only {cur} tokens (gap)")
            elif speculative:
                handle.rollback()  # drop any uncommitted previous tree
            if (self._use_graphs and T == 1 and not speculative
                    and prompts is None and position_ids is None
                    and tree_mask is None and adapter is None
                    and start_pos == cur and handle.pos_offset == 0):
                return self._graphed_decode(session_id, handle, h, start_pos)
            pos = (position_ids.to(self.device).int()
                   if position_ids is not None else None)
            tm = tree_mask.to(self.device) if tree_mask is not None else None
            # long-prefill sequence chunking (ref backend.py:525-531,
            # 839-845 max_chunk_size_bytes): bound activation memory by
            # running the span over sequence slices; each chunk's KV lands
            # in the pages before the next chunk attends (the prefill
            # kernel's continuation path)
            if (tm is None and pos is None and not speculative
                    and T > self.max_chunk_tokens):
                from bloombee_amd.utils.logging import debug_log
                debug_log("inference", logger,
                          "chunking prefill T=%d by %d", T,
                          self.max_chunk_tokens)
                outs = []
                for t0 in range(0, T, self.max_chunk_tokens):
                    t1 = min(T, t0 + self.max_chunk_tokens)
                    handle.extend(t1 - t0)
                    sp = torch.full((B,), start_pos + t0, dtype=torch.int32,
                                    device=self.device)
                    outs.append(self.stack.forward_inference(
                        h[:, t0:t1], handle, sp, deep_prompts=deep))
                out = torch.cat(outs, dim=1)
            else:
                sp = torch.full((B,), start_pos, dtype=torch.int32,
                                device=self.device)
                handle.extend(T, speculative=speculative)
                out = self.stack.forward_inference(h, handle, sp, pos,
                                                   tree_mask=tm,
                                                   deep_prompts=deep)
            from bloombee_amd.utils import activation_dumper
            if activation_dumper.enabled():
                activation_dumper.capture_activation(
                    f"blocks{self.start}_{self.end}", out)
            return out

        return self.pool.submit(run, PRIORITY_INFERENCE).result()

    def _graphed_decode(self, session_id: str, handle, h: torch.Tensor,
                        start_pos: int) -> torch.Tensor:
        """Plain decode step (T=1, committed position) as a hipGraph replay.

        Runs ON the worker thread. All host-varying inputs are copied into
        persistent device buffers before replay; kv growth (extend +
        page-table flush) happens outside the graph, exactly like the
        single-rank pipeline capture (parallel/pipeline.py:151-175).
        capture_error_mode="thread_local" keeps the channels pump (RCCL ops
        on other threads) legal during capture."""
        B = h.shape[0]
        st = self._graphs.get(session_id)
        if st is None or st["B"] != B:
            st = {"B": B,
                  "in": torch.empty_like(h),
                  "pos": torch.empty(B, dtype=torch.int32,
                                     device=self.device),
                  "out": None, "graph": None, "warm": 0}
            self._graphs[session_id] = st
        st["in"].copy_(h)
        st["pos"].fill_(start_pos)
        handle.extend(1)
        handle.page_table()
        if st["graph"] is None:
            if st["warm"] < 2:  # warm the allocator/kernels before capture
                st["warm"] += 1
                return self.stack.forward_inference(st["in"], handle,
                                                    st["pos"])
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, capture_error_mode="thread_local"):
                st["out"] = self.stack.forward_inference(st["in"], handle,
                                                         st["pos"])
            st["graph"] = g
            self.graphs_captured += 1
            logger.info("decode step blocks[%d:%d] B=%d captured as hipGraph",
                        self.start, self.end, B)
        st["graph"].replay()
        return st["out"]

    def prune_tree(self, hidden: torch.Tensor, tokens: list,
                   parents: list):
        """Last-block mid-network pruning (ref backend.py:763-777): returns
        (kept_hidden, keep_indices) or (hidden, None) when disabled."""
        if self.pruner is None or hidden.shape[0] != 1:
            return hidden, None

        def run():
            keep = self.pruner.keep_indices(hidden[0], tokens, parents)
            return hidden[:, keep], keep

        return self.pool.submit(run, PRIORITY_INFERENCE).result()

    def spec_commit(self, session_id: str, keep: list) -> None:
        """Accept tree nodes: compact their KV + commit (see paged.py)."""
        handle = self._session(session_id).handle
        self.pool.submit(lambda: handle.reorder_and_commit(keep),
                         PRIORITY_INFERENCE).result()

    def forward(self, hidden: torch.Tensor,
                prompts: Optional[torch.Tensor] = None,
                adapter: Optional[str] = None) -> torch.Tensor:
        """Training-path forward (no KV cache, full sequence). prompts:
        optional deep p-tune tensor (n_local_blocks, pre, H) added to the
        prompt positions at every block input (ref client/ptune.py deep)."""

        def run():
            from bloombee_amd.utils.peft import using_adapter
            h = hidden.to(self.device).to(self.config.dtype)
            dp = prompts.to(self.device) if prompts is not None else None
            with torch.no_grad(), using_adapter(adapter):
                return self.stack.forward_train(h, deep_prompts=dp)

        return self.pool.submit(run, PRIORITY_TRAIN).result()

    def backward(self, hidden_in: torch.Tensor, grad_out: torch.Tensor,
                 prompts: Optional[torch.Tensor] = None,
                 adapter: Optional[str] = None,
                 ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
        """Re-forward + backward; returns (grad wrt span input, grad wrt deep
        prompts or None). Server weights are frozen — only input/prompt grads
        flow (ref backend.py:106-109, 427-462)."""

        def run():
            from bloombee_amd.utils.peft import using_adapter
            h = hidden_in.to(self.device).to(self.config.dtype)
            h = h.detach().requires_grad_(True)
            dp = None
            if prompts is not None:
                dp = prompts.to(self.device).detach().requires_grad_(True)
            with torch.enable_grad(), using_adapter(adapter):
                out = self.stack.forward_train(h, deep_prompts=dp)
                g = grad_out.to(out.device).to(out.dtype)
                leaves = (h,) if dp is None else (h, dp)
                grads = torch.autograd.grad(out, leaves, g)
            return (grads[0], grads[1] if dp is not None else None)

        return self.pool.submit(run, PRIORITY_TRAIN).result()

    # -- info -------------------------------------------------------------
    def info(self) -> dict:
        return {
            "model_type": self.config.model_type,
            "start_block": self.start,
            "end_block": self.end,
            "torch_dtype": self.config.torch_dtype,
            "cache_tokens_left": self.kv_pool.tokens_left,
            "num_sessions": len(self.sessions),
        }

    def shutdown(self):
        with self._lock:
            for state in self.sessions.values():
                state.handle.close()
            self.sessions.clear()
        self.pool.shutdown()
