"""Pipeline-parallel decode over torch.distributed (RCCL on GPU, gloo on CPU).

This is the single-node fast path of the reference's inter-server pipeline:
on one 8xMI355X node the client->S1->...->SN->client activation hops become
RCCL send/recv over xGMI (SURVEY.md §2.7 "Pipeline parallelism": xGMI is 7
point-to-point links per GPU, so neighbor-only PP traffic rides one dedicated
link), and the reference's server->server rpc_push + push-only-downstream
decode (handler.py:2239-2760, inference_session.py:178-196) becomes
micro-batch pipelining: stage r computes micro-batch j while j-1 is in
flight to stage r+1.

Topology: rank r hosts the contiguous block range [r*L/N, (r+1)*L/N); rank 0
additionally holds the client role (embeddings, final norm, LM head) —
matching the reference client split (client/lm_head.py, model.py:80-118).
The ring is r -> r+1 -> ... -> N-1 -> 0.
"""
from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F

from bloombee_amd.engine import BlockStack
from bloombee_amd.kv.paged import PagedKVCache
from bloombee_amd.kv.views import SessionView
from bloombee_amd.models.base import ModelConfig, resolve_config
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


def layer_range(num_layers: int, rank: int, world: int):
    per = num_layers // world
    extra = num_layers % world
    start = rank * per + min(rank, extra)
    end = start + per + (1 if rank < extra else 0)
    return start, end


class PipelineStage:
    """One rank of the pipelined decoder."""

    def __init__(self, config_or_name, device, global_batch: int,
                 micro_batches: int = 0, seed: int = 0,
                 kv_max_tokens: int = 1 << 17, max_session_len: int = 4096,
                 tp: int = 1, tp_mode: str = "tensor"):
        cfg = (config_or_name if isinstance(config_or_name, ModelConfig)
               else resolve_config(config_or_name))
        self.config = cfg
        self.device = torch.device(device)
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        self.world = dist.get_world_size() if dist.is_initialized() else 1
        assert self.world % tp == 0, f"world {self.world} not divisible by tp {tp}"
        self.tp = tp
        self.pp_world = self.world // tp
        self.pp_stage = self.rank // tp
        self.tp_rank = self.rank % tp
        # per-stage tensor-parallel groups (RCCL all-reduce over xGMI inside
        # a worker — BASELINE.json config 4; created on every rank)
        self.tp_group = None
        if tp > 1:
            for st in range(self.pp_world):
                g = dist.new_group(list(range(st * tp, (st + 1) * tp)))
                if st == self.pp_stage:
                    self.tp_group = g
        self.global_batch = global_batch
        if micro_batches <= 0:
            # enough micro-batches to overlap the stage chain, but keep each
            # slice >= 8 sequences: BW-bound decode kernels lose ~2x at B8
            # vs B32 (benchmarks/bench_kernels.py attn_decode)
            if self.pp_world > 1:
                micro_batches = max(1, min(2 * self.pp_world,
                                           global_batch // 8))
            else:
                micro_batches = 1
        while global_batch % micro_batches != 0:
            micro_batches -= 1
        self.M = micro_batches
        self.mb = global_batch // self.M

        start, end = layer_range(cfg.num_hidden_layers, self.pp_stage,
                                 self.pp_world)
        logger.info(f"rank {self.rank}/{self.world} (pp {self.pp_stage} tp "
                    f"{self.tp_rank}): layers [{start}, {end})")
        self.tp_mode = tp_mode
        if tp > 1 and tp_mode == "context":
            # context parallelism: full weights on every rank; PREFILL
            # shards the sequence with ring attention (cp_prefill_llama)
            # and all-gathers K/V so decode replicates locally. Composes
            # with pp: each stage's cp group ring-prefills its own layer
            # range, full hidden flows leader-to-leader between stages.
            self.stack = BlockStack(cfg, start, end, device=self.device,
                                    seed=seed)
        elif tp > 1 and tp_mode == "expert":
            # expert parallelism inside a stage (BASELINE config 4 at scale;
            # beyond the reference, which runs all experts locally): every
            # rank keeps the full attention/router weights and ONLY its
            # expert shard; each block's partial MoE output crosses the
            # group with one RCCL all-reduce (parallel/expert.py). Non-MoE
            # blocks compute replicated.
            from bloombee_amd.parallel.expert import shard_experts
            self.stack = BlockStack(cfg, start, end, device=self.device,
                                    seed=seed)
            n_moe = 0
            for blk in self.stack.blocks:
                if hasattr(blk, "expert_gate_up_w"):
                    shard_experts(blk, self.tp_rank, tp, group=self.tp_group)
                    n_moe += 1
            logger.info("EP group %d-way: %d MoE blocks sharded", tp, n_moe)
        elif tp > 1:
            from bloombee_amd.parallel.tensor import TPBlockStack
            self.stack = TPBlockStack(cfg, start, end, device=self.device,
                                      seed=seed, group=self.tp_group)
        else:
            self.stack = BlockStack(cfg, start, end, device=self.device, seed=seed)
        self.kv_pool = self.stack.make_kv(kv_max_tokens)
        self.kv = self.kv_pool.allocate(global_batch, max_session_len)

        self.is_client = self.rank == 0
        if self.is_client:
            gen = torch.Generator().manual_seed(seed)
            dt = cfg.dtype
            self.embed = (torch.randn(cfg.vocab_size, cfg.hidden_size, generator=gen)
                          .mul_(0.02).to(dt).to(self.device))
            self.final_norm_w = torch.ones(cfg.hidden_size, dtype=dt, device=self.device)
            self.lm_head_w = (self.embed if cfg.tie_word_embeddings else
                              (torch.randn(cfg.vocab_size, cfg.hidden_size, generator=gen)
                               .mul_(0.02).to(dt).to(self.device)))
        # inter-stage chain runs between tp leaders (tp_rank 0); activations
        # are replicated within a stage after each block's all-reduce
        self.next_rank = ((self.pp_stage + 1) % self.pp_world) * tp
        self.prev_rank = ((self.pp_stage - 1) % self.pp_world) * tp
        self.is_leader = self.tp_rank == 0
        # persistent recv buffers per micro-batch
        self._recv_buf = [torch.empty(self.mb, 1, cfg.hidden_size, dtype=cfg.dtype,
                                      device=self.device) for _ in range(self.M)]
        # hipGraph decode capture (single-rank fast path): the whole decode
        # step — embed, 32 blocks, LM head, argmax, position bump — replays as
        # one graph (reference utils/cuda_graphs.py analog; on MI355X the
        # ~300-launch step is otherwise launch-gap bound)
        from bloombee_amd.config import get_config
        self._use_graphs = (get_config().use_hip_graphs and self.world == 1
                            and self.device.type == "cuda")
        self._graph = None
        self._ids_buf: Optional[torch.Tensor] = None
        self._pos_buf: Optional[torch.Tensor] = None
        self._eager_steps = 0

    # ------------------------------------------------------------------
    def _run_local(self, hidden: torch.Tensor, view, start_pos: torch.Tensor):
        return self.stack.forward_inference(hidden, view, start_pos)

    def _views(self):
        return [SessionView(self.kv, j * self.mb, (j + 1) * self.mb)
                for j in range(self.M)]

    # ------------------------------------------------------------------
    # single-rank graphed decode
    # ------------------------------------------------------------------
    def _client_step_local(self):
        """Whole decode step on persistent buffers (graph-capturable: every
        host-varying quantity lives in a device tensor)."""
        B = self.global_batch
        hid = F.embedding(self._ids_buf.view(B, 1), self.embed)
        h = self.stack.forward_inference(hid, self.kv, self._pos_buf)
        from bloombee_amd import ops

        y = ops.rms_norm(h[:, -1], self.final_norm_w, self.config.rms_norm_eps)
        # vocab projection stays on hipBLASLt: at N=128k it streams the
        # 1.05 GB weight at 5.5 TB/s vs 5.3 for the skinny kernel (measured
        # — the skinny kernel only wins the small-N decode shapes)
        nxt = F.linear(y, self.lm_head_w).float().argmax(-1)
        self._ids_buf.copy_(nxt)
        self._pos_buf += 1

    def _decode_round_single(self, ids: torch.Tensor) -> torch.Tensor:
        B = self.global_batch
        if self._ids_buf is None:
            self._ids_buf = torch.empty(B, dtype=torch.long, device=self.device)
            self._pos_buf = torch.empty(B, dtype=torch.int32, device=self.device)
            self._pos_buf.fill_(self.kv.seqs[0].l_spec)
        if ids is not self._ids_buf:
            self._ids_buf.copy_(ids.to(self.device))
        self.kv.extend(1)
        self.kv.page_table()  # flush any new pages to the device table
        if not self._use_graphs:
            self._client_step_local()
            return self._ids_buf
        if self._graph is None:
            if self._eager_steps < 2:  # warm up allocator/kernels first
                self._eager_steps += 1
                self._client_step_local()
                return self._ids_buf
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._client_step_local()
            self._graph = g
            logger.info("decode step captured as hipGraph")
        self._graph.replay()
        return self._ids_buf

    @torch.no_grad()
    def decode_round(self, ids: Optional[torch.Tensor]) -> Optional[torch.Tensor]:
        """One decode step for the whole global batch, micro-batch pipelined.

        rank 0: `ids` is (global_batch,) current tokens; returns next tokens.
        other ranks: pass None; returns None.
        """
        if self.world == 1 and self.device.type == "cuda":
            return self._decode_round_single(ids)
        cfg = self.config
        B, M, mb = self.global_batch, self.M, self.mb
        pos0 = self.kv.seqs[0].l_spec  # all sequences advance in lockstep
        start_all = torch.full((B,), pos0, dtype=torch.int32, device=self.device)
        self.kv.extend(1)
        views = self._views()
        pending = []

        if self.pp_stage == 0:
            if self.is_client:
                hid = F.embedding(ids.view(B, 1).to(self.device), self.embed)
            else:
                hid = torch.empty(B, 1, cfg.hidden_size, dtype=cfg.dtype,
                                  device=self.device)
            if self.tp > 1:
                dist.broadcast(hid, src=0, group=self.tp_group)
            outs: List[torch.Tensor] = []
            for j in range(M):
                h = self._run_local(hid[j * mb:(j + 1) * mb], views[j],
                                    start_all[j * mb:(j + 1) * mb])
                if self.pp_world > 1 and self.is_leader:
                    pending.append(dist.isend(h.contiguous(), self.next_rank))
                elif self.pp_world == 1:
                    outs.append(h)
            if self.pp_world > 1 and self.is_client:
                for j in range(M):
                    dist.recv(self._recv_buf[j], self.prev_rank)
                    outs.append(self._recv_buf[j])
            for w in pending:
                w.wait()
            if not self.is_client:
                return None
            hidden = torch.cat(outs, dim=0)
            return self._lm_head(hidden[:, -1])
        else:
            for j in range(M):
                if self.is_leader:
                    dist.recv(self._recv_buf[j], self.prev_rank)
                if self.tp > 1:
                    dist.broadcast(self._recv_buf[j],
                                   src=self.pp_stage * self.tp,
                                   group=self.tp_group)
                h = self._run_local(self._recv_buf[j], views[j],
                                    start_all[j * mb:(j + 1) * mb])
                if self.is_leader:
                    pending.append(dist.isend(h.contiguous(), self.next_rank))
            for w in pending:
                w.wait()
            return None

    @torch.no_grad()
    def _prefill_cp(self, ids: Optional[torch.Tensor], T: int):
        """Context-parallel one-shot prefill, composable with pp: each
        stage's cp group shards the SEQUENCE, runs ring attention over the
        stage's layers, and all-gathers K/V into every rank's pool
        (parallel/sequence.cp_prefill_llama); the full hidden flows
        leader-to-leader between stages like the generic prefill."""
        from bloombee_amd.parallel.sequence import (cp_prefill_gemma4,
                                                    cp_prefill_llama)

        cfg = self.config
        B = self.global_batch
        assert self.kv.seqs[0].l_spec == 0, "CP prefill is one-shot"
        assert T % self.tp == 0, "prompt length must divide by cp degree"
        hid = torch.empty(B, T, cfg.hidden_size, dtype=cfg.dtype,
                          device=self.device)
        if self.pp_stage == 0:
            if self.is_client:
                hid = F.embedding(ids.to(self.device), self.embed)
        elif self.is_leader:
            dist.recv(hid, self.prev_rank)
        if self.tp > 1 and dist.is_initialized():
            dist.broadcast(hid, src=self.pp_stage * self.tp,
                           group=self.tp_group)
        Tl = T // self.tp
        self.kv.extend(T)
        shard = hid[:, self.tp_rank * Tl:(self.tp_rank + 1) * Tl].contiguous()
        cp_fn = (cp_prefill_gemma4
                 if hasattr(self.stack.blocks[0], "pre_ffn_norm_w")
                 else cp_prefill_llama)
        out_shard = cp_fn(self.stack, self.kv, shard,
                          self.tp_rank, self.tp, self.tp_group)
        if self.tp > 1 and dist.is_initialized():
            outs = [torch.empty_like(out_shard) for _ in range(self.tp)]
            dist.all_gather(outs, out_shard.contiguous(),
                            group=self.tp_group)
            hidden = torch.cat(outs, dim=1)
        else:
            hidden = out_shard
        if self.pp_world > 1:
            if self.is_leader:
                dist.send(hidden.contiguous(), self.next_rank)
            if self.is_client:
                final = torch.empty_like(hid)
                dist.recv(final, self.prev_rank)
                return self._lm_head(final[:, -1])
            return None
        if not self.is_client:
            return None
        return self._lm_head(hidden[:, -1])

    @torch.no_grad()
    def prefill_round(self, ids: Optional[torch.Tensor], T: int) -> Optional[torch.Tensor]:
        """Prefill T prompt tokens (chunked by micro-batch over the batch dim)."""
        if self.tp > 1 and self.tp_mode == "context":
            return self._prefill_cp(ids, T)
        cfg = self.config
        B, M, mb = self.global_batch, self.M, self.mb
        pos0 = self.kv.seqs[0].l_spec
        start_all = torch.full((B,), pos0, dtype=torch.int32, device=self.device)
        self.kv.extend(T)
        views = self._views()
        pending = []
        bufs = [torch.empty(mb, T, cfg.hidden_size, dtype=cfg.dtype,
                            device=self.device) for _ in range(M)]
        if self.pp_stage == 0:
            if self.is_client:
                hid = F.embedding(ids.to(self.device), self.embed)
            else:
                hid = torch.empty(B, T, cfg.hidden_size, dtype=cfg.dtype,
                                  device=self.device)
            if self.tp > 1:
                dist.broadcast(hid, src=0, group=self.tp_group)
            outs = []
            for j in range(M):
                h = self._run_local(hid[j * mb:(j + 1) * mb], views[j],
                                    start_all[j * mb:(j + 1) * mb])
                if self.pp_world > 1 and self.is_leader:
                    pending.append(dist.isend(h.contiguous(), self.next_rank))
                elif self.pp_world == 1:
                    outs.append(h)
            if self.pp_world > 1 and self.is_client:
                for j in range(M):
                    dist.recv(bufs[j], self.prev_rank)
                    outs.append(bufs[j])
            for w in pending:
                w.wait()
            if not self.is_client:
                return None
            hidden = torch.cat(outs, dim=0)
            return self._lm_head(hidden[:, -1])
        else:
            for j in range(M):
                if self.is_leader:
                    dist.recv(bufs[j], self.prev_rank)
                if self.tp > 1:
                    dist.broadcast(bufs[j], src=self.pp_stage * self.tp,
                                   group=self.tp_group)
                h = self._run_local(bufs[j], views[j], start_all[j * mb:(j + 1) * mb])
                if self.is_leader:
                    pending.append(dist.isend(h.contiguous(), self.next_rank))
            for w in pending:
                w.wait()
            return None

    def _lm_head(self, hidden_last: torch.Tensor) -> torch.Tensor:
        from bloombee_amd import ops

        y = ops.rms_norm(hidden_last, self.final_norm_w, self.config.rms_norm_eps)
        return F.linear(y, self.lm_head_w).float().argmax(-1)


def init_distributed(device_type: str = "auto") -> str:
    """Initialize torch.distributed from torchrun env (one process per GPU,
    RCCL over xGMI; gloo on CPU-only hosts/tests). Returns device string."""
    if "RANK" not in os.environ:
        return "cuda:0" if torch.cuda.is_available() else "cpu"
    rank = int(os.environ["RANK"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    use_gpu = torch.cuda.is_available() and device_type != "cpu"
    backend = "nccl" if use_gpu else "gloo"
    if use_gpu:
        torch.cuda.set_device(local_rank)
    dist.init_process_group(backend=backend)
    return f"cuda:{local_rank}" if use_gpu else "cpu"
