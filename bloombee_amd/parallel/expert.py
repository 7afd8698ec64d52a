"""Expert parallelism for MoE blocks (beyond the reference, which runs all
experts locally — models/mixtral/block.py:13-137): experts are partitioned
across the ranks of a torch.distributed group, each rank computes only its
local experts' contributions for the (replicated) token batch, and the
partial outputs are summed with one bucketed all-reduce — RCCL over xGMI on
an MI355X node, gloo in CPU tests. With top-k=2 routing each token row has
at most two contributions, so the cross-rank sum is bitwise identical to
the single-process expert loop.

Design note: dispatch-style EP (all-to-all of routed tokens) sends less
data when the activation batch is large, but at decode batch sizes the
all-reduce of one (B,H) tensor per layer is smaller than two all-to-alls
plus index traffic, and it reuses the TP all-reduce path (xGMI ring,
per-link bound) — see parallel/tensor.py.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist


def shard_experts(block, rank: int, world: int,
                  group: Optional[object] = None):
    """Keep only this rank's expert slice of a MixtralBlock (call after
    weight init so every rank draws the identical full tensors first).
    The router stays replicated — identical logits on every rank."""
    E = block.E
    per = (E + world - 1) // world
    e0, e1 = rank * per, min(E, (rank + 1) * per)
    block.expert_gate_up_w = torch.nn.Parameter(
        block.expert_gate_up_w.data[e0:e1].clone(), requires_grad=False)
    block.expert_down_w = torch.nn.Parameter(
        block.expert_down_w.data[e0:e1].clone(), requires_grad=False)
    block.ep_range = (e0, e1)
    block.ep_group = group
    block.ep_enabled = True
    return block


def ep_all_reduce(t: torch.Tensor, group: Optional[object]) -> torch.Tensor:
    """Sum partial expert outputs across the EP group. group=None means the
    DEFAULT process group (never 'skip' — see parallel/tensor.py)."""
    if dist.is_available() and dist.is_initialized():
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=group)
    return t
