"""Which contiguous block range should a joining server host?

Port of the reference's semantics (server/block_selection.py:23-95): pick the
start of the window whose worst-covered block has the lowest aggregate
throughput; rebalance when moving would improve the swarm's bottleneck by
more than `balance_quality`.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

from bloombee_amd.data_structures import RemoteModuleInfo, ServerState


def block_throughputs(infos: Sequence[RemoteModuleInfo]) -> List[float]:
    out = []
    for info in infos:
        tp = sum(s.throughput for s in info.servers.values()
                 if s.state == ServerState.ONLINE)
        out.append(tp)
    return out


def choose_best_blocks(num_blocks: int, infos: Sequence[RemoteModuleInfo],
                       ) -> List[int]:
    """Indices [start, start+num_blocks) minimizing the window's coverage."""
    tp = block_throughputs(infos)
    n = len(tp)
    num_blocks = min(num_blocks, n)
    best_start, best_key = 0, None
    for s in range(0, n - num_blocks + 1):
        window = sorted(tp[s:s + num_blocks])
        key = window  # lexicographic: worst block first (ref min-window search)
        if best_key is None or key < best_key:
            best_start, best_key = s, key
    return list(range(best_start, best_start + num_blocks))


def should_choose_other_blocks(my_peer_id: str, infos: Sequence[RemoteModuleInfo],
                               balance_quality: float = 0.75) -> bool:
    """True if re-running selection without us would improve the swarm
    bottleneck by enough to justify a move (ref block_selection.py:39-95)."""
    tp = block_throughputs(infos)
    if not tp:
        return False
    cur_bottleneck = min(tp)
    # simulate: remove our contribution, re-place ourselves optimally
    without = []
    my_len = 0
    my_tp = 0.0
    for info, t in zip(infos, tp):
        mine = info.servers.get(my_peer_id)
        if mine is not None and mine.state == ServerState.ONLINE:
            t -= mine.throughput
            my_len += 1
            my_tp = mine.throughput
        without.append(t)
    if my_len == 0:
        return False
    best = choose_best_blocks(my_len, [
        RemoteModuleInfo(uid=i.uid, servers={
            k: v for k, v in i.servers.items() if k != my_peer_id})
        for i in infos])
    after = list(without)
    for b in best:
        after[b] += my_tp
    new_bottleneck = min(after)
    return new_bottleneck > cur_bottleneck / balance_quality
