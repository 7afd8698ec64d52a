"""Intra-worker tensor parallelism over RCCL/xGMI.

Parity target: reference server/flexgen_tensor_parallel.py (828 LoC) —
column-split QKV + gate/up, row-split O + down, partial-sum reduction.
MI355X-native redesign: instead of the reference's
single-process `torch.cuda.comm.reduce_add` over device lists, shards are
one-rank-per-GPU processes in a torch.distributed group; the two partial-sum
reductions per block are RCCL all-reduces over xGMI (backend "nccl" IS RCCL
on ROCm; gloo on CPU for tests).

Sharding (llama-family block):
  * q/k/v heads split across ranks (Hq % tp == 0, Hkv % tp == 0): each rank
    holds Hq/tp query + Hkv/tp kv heads and ITS OWN paged KV pool shard —
    the KV cache is sharded for free.
  * o_w column-split to match the local q heads; partials all-reduced.
  * gate/up row-split, down column-split; partials all-reduced (fused with
    the residual on rank-identical values).
Norms and residuals are replicated (cheap, avoids a third collective).
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from bloombee_amd import ops
from bloombee_amd.engine import BlockStack
from bloombee_amd.kv.paged import PagedKVCache, SessionHandle
from bloombee_amd.models.base import ModelConfig
from bloombee_amd.models.llama.block import LlamaBlock
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


def _all_reduce(t: torch.Tensor, group) -> torch.Tensor:
    # group=None means the DEFAULT process group, not "no group"
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return t
    tf = t.float()
    dist.all_reduce(tf, op=dist.ReduceOp.SUM, group=group)
    return tf.to(t.dtype)


class TPShardedLlamaBlock(torch.nn.Module):
    """One llama block's shard on this rank. Built by slicing a fully
    materialized block (all ranks draw identical random weights, then keep
    their slice — deterministic across world sizes)."""

    def __init__(self, full: LlamaBlock, tp_rank: int, tp_world: int, group):
        super().__init__()
        cfg = full.config
        Hq, Hkv, D, I = full.Hq, full.Hkv, full.D, full.I
        assert Hq % tp_world == 0 and Hkv % tp_world == 0 and I % tp_world == 0, \
            f"heads/intermediate not divisible by tp={tp_world}"
        self.config = cfg
        self.layer_index = full.layer_index
        self.group = group
        self.tp_rank, self.tp_world = tp_rank, tp_world
        self.Hq, self.Hkv = Hq // tp_world, Hkv // tp_world
        self.D, self.I = D, I // tp_world
        self.scale = full.scale
        self.rope = full.rope

        qs = slice(tp_rank * self.Hq * D, (tp_rank + 1) * self.Hq * D)
        ks = slice(Hq * D + tp_rank * self.Hkv * D,
                   Hq * D + (tp_rank + 1) * self.Hkv * D)
        vs = slice((Hq + Hkv) * D + tp_rank * self.Hkv * D,
                   (Hq + Hkv) * D + (tp_rank + 1) * self.Hkv * D)

        def p(t):
            return torch.nn.Parameter(t.detach().clone(), requires_grad=False)

        self.input_norm_w = p(full.input_norm_w)
        self.qkv_w = p(torch.cat([full.qkv_w[qs], full.qkv_w[ks],
                                  full.qkv_w[vs]], dim=0))
        self.o_w = p(full.o_w[:, qs])
        self.post_norm_w = p(full.post_norm_w)
        gsl = slice(tp_rank * self.I, (tp_rank + 1) * self.I)
        usl = slice(I + tp_rank * self.I, I + (tp_rank + 1) * self.I)
        self.gate_up_w = p(torch.cat([full.gate_up_w[gsl],
                                      full.gate_up_w[usl]], dim=0))
        self.down_w = p(full.down_w[:, gsl])

    @torch.no_grad()
    def forward_inference(self, hidden: torch.Tensor, kv: SessionHandle,
                          start_pos: torch.Tensor,
                          position_ids: Optional[torch.Tensor] = None,
                          ) -> torch.Tensor:
        B, T, H = hidden.shape
        Hq, Hkv, D = self.Hq, self.Hkv, self.D
        cfg = self.config
        x = ops.rms_norm(hidden, self.input_norm_w, cfg.rms_norm_eps)
        qkv = ops.linear(x, self.qkv_w)
        cos, sin = self.rope.get(hidden.device)
        kp = kv.k_pages(self.layer_index)
        vp = kv.v_pages(self.layer_index)
        pt = kv.page_table()
        ops.rope_kv_write_(qkv, Hq, Hkv, cos, sin, position_ids, kp, vp, pt,
                           start_pos)
        attn = ops.attn_paged_qkv(qkv, Hq, Hkv, kp, vp, pt, start_pos, self.scale)
        a = ops.linear(attn, self.o_w)
        a = _all_reduce(a, self.group)            # RCCL all-reduce #1
        h2, y = ops.rms_norm_residual(a, hidden, self.post_norm_w, cfg.rms_norm_eps)
        m = ops.linear(ops.swiglu(ops.linear(y, self.gate_up_w)), self.down_w)
        m = _all_reduce(m, self.group)            # RCCL all-reduce #2
        return h2 + m

    def forward(self, *a, **k):
        return self.forward_inference(*a, **k)


class TPBlockStack:
    """A BlockStack sharded across the ranks of `group` (llama family)."""

    def __init__(self, config: ModelConfig, start: int, end: int,
                 device="cpu", seed: int = 0, group=None):
        self.config = config
        self.start, self.end = start, end
        self.device = torch.device(device)
        self.group = group
        tp_world = dist.get_world_size(group) if dist.is_initialized() else 1
        tp_rank = dist.get_rank(group) if dist.is_initialized() else 0
        self.tp_world, self.tp_rank = tp_world, tp_rank
        full = BlockStack(config, start, end, device=device, seed=seed)
        self.blocks = torch.nn.ModuleList([
            TPShardedLlamaBlock(b, tp_rank, tp_world, group)
            for b in full.blocks])
        del full

    def make_kv(self, max_tokens: int) -> PagedKVCache:
        return PagedKVCache(
            num_layers=len(self.blocks),
            num_kv_heads=self.config.num_key_value_heads // self.tp_world,
            head_dim=self.config.head_dim,
            max_tokens=max_tokens,
            device=self.device,
            dtype=self.config.dtype,
        )

    @torch.no_grad()
    def forward_inference(self, hidden, kv, start_pos, position_ids=None,
                          tree_mask=None):
        for blk in self.blocks:
            hidden = blk.forward_inference(hidden, kv, start_pos, position_ids)
        return hidden
