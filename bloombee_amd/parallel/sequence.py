"""Ring-attention context parallelism (beyond the reference — SURVEY §2.7
marks sequence/context parallel ABSENT there, with xGMI ring attention noted
as the natural MI355X extension): the sequence is sharded across the ranks
of a group, each rank holds Q/K/V for its contiguous shard, and K/V blocks
rotate around the ring (isend/irecv pairs — point-to-point xGMI traffic,
one dedicated link per neighbor) while every rank folds each arriving block
into its online-softmax state. Causality: rank r folds only source shards
<= r, with the triangular mask applied on the diagonal shard.

The fold is the flash-attention (m, l, acc) merge in fp32, so the result
matches single-device attention to numerical tolerance regardless of ring
size.
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist


def _fold(state, s, v, row_pos=None, col_pos=None):
    """Fold one score block into the online-softmax state.

    state: (m, l, acc) with m,l (B,H,T,1) and acc (B,H,T,D), all fp32.
    s: (B,H,T,S) scores; v: (B,H,S,D). If row_pos/col_pos given, mask
    s[t, c] where col_pos[c] > row_pos[t] (causal)."""
    m, l, acc = state
    if row_pos is not None:
        dead = col_pos.view(1, 1, 1, -1) > row_pos.view(1, 1, -1, 1)
        s = s.masked_fill(dead, float("-inf"))
    m_new = torch.maximum(m, s.amax(-1, keepdim=True))
    # fully-masked rows keep m = -inf; exp(-inf - -inf) guards below
    m_safe = torch.where(torch.isinf(m_new), torch.zeros_like(m_new), m_new)
    p = torch.exp(s - m_safe)
    if row_pos is not None:
        p = p.masked_fill(dead, 0.0)
    corr = torch.exp(torch.where(torch.isinf(m), m_new.new_zeros(()), m - m_safe))
    corr = torch.where(torch.isinf(m), torch.zeros_like(corr), corr)
    l = l * corr + p.sum(-1, keepdim=True)
    acc = acc * corr + p @ v
    return (m_new, l, acc)


def _ring_exchange(t: torch.Tensor, rank: int, world: int,
                   group=None) -> torch.Tensor:
    """Send t to (rank+1) % world, receive the previous rank's tensor."""
    out = torch.empty_like(t)
    send = dist.isend(t.contiguous(), (rank + 1) % world, group=group)
    recv = dist.irecv(out, (rank - 1) % world, group=group)
    send.wait()
    recv.wait()
    return out


def ring_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   rank: int, world: int, group=None,
                   scale: Optional[float] = None) -> torch.Tensor:
    """Causal attention over a sequence sharded as contiguous T_loc blocks.

    q/k/v: (B, H, T_loc, D) — this rank's shard (global positions
    [rank*T_loc, (rank+1)*T_loc)). Returns (B, H, T_loc, D)."""
    B, H, T, D = q.shape
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    qf = q.float()
    state = (torch.full((B, H, T, 1), float("-inf")),
             torch.zeros(B, H, T, 1),
             torch.zeros(B, H, T, D))
    row_pos = torch.arange(rank * T, (rank + 1) * T)
    k_cur, v_cur = k.float(), v.float()
    src = rank
    for step in range(world):
        if src < rank:
            s = (qf @ k_cur.transpose(-1, -2)) * scale
            state = _fold(state, s, v_cur)
        elif src == rank:
            s = (qf @ k_cur.transpose(-1, -2)) * scale
            col_pos = torch.arange(src * T, (src + 1) * T)
            state = _fold(state, s, v_cur, row_pos, col_pos)
        # else: future shard — causally invisible, just keep rotating
        if step < world - 1:
            k_cur = _ring_exchange(k_cur, rank, world, group)
            v_cur = _ring_exchange(v_cur, rank, world, group)
            src = (src - 1) % world
    m, l, acc = state
    return (acc / l.clamp_min(1e-30)).to(q.dtype)


def ring_attention_local(q, k, v, world: int,
                         scale: Optional[float] = None) -> torch.Tensor:
    """Single-process simulation of the ring over `world` shards (unit-tests
    the fold/rotation logic at any depth without process groups).

    q/k/v: (B, H, T_total, D) with world | T_total. Returns full output."""
    B, H, T, D = q.shape
    Tl = T // world
    outs = []
    for r in range(world):
        qr = q[:, :, r * Tl:(r + 1) * Tl].float()
        if scale is None:
            sc = 1.0 / math.sqrt(D)
        else:
            sc = scale
        state = (torch.full((B, H, Tl, 1), float("-inf")),
                 torch.zeros(B, H, Tl, 1),
                 torch.zeros(B, H, Tl, D))
        row_pos = torch.arange(r * Tl, (r + 1) * Tl)
        # same visitation order as the ring: own shard first, then r-1, ...
        for src in [r] + [(r - 1 - i) % world for i in range(world - 1)]:
            if src > r:
                continue
            kc = k[:, :, src * Tl:(src + 1) * Tl].float()
            vc = v[:, :, src * Tl:(src + 1) * Tl].float()
            s = (qr @ kc.transpose(-1, -2)) * sc
            if src == r:
                col_pos = torch.arange(src * Tl, (src + 1) * Tl)
                state = _fold(state, s, vc, row_pos, col_pos)
            else:
                state = _fold(state, s, vc)
        m, l, acc = state
        outs.append((acc / l.clamp_min(1e-30)).to(q.dtype))
    return torch.cat(outs, dim=2)
