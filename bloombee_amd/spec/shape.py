"""Tree-shape planning (parity: reference spec_decoding_tree_shape.py:74-250 —
Sequoia-style expected-accepted-length maximization under a node budget,
driven by an online acceptance histogram)."""
from __future__ import annotations

from typing import Dict, List

import math


class AcceptanceStats:
    """Online per-depth acceptance-rate histogram (ref acceptance histogram)."""

    def __init__(self, max_depth: int = 16, prior: float = 0.6):
        self.max_depth = max_depth
        self.accepted = [1.0] * max_depth   # laplace-ish prior
        self.offered = [1.0 / max(prior, 1e-3)] * max_depth

    def record(self, accepted_len: int, offered_depth: int) -> None:
        for d in range(min(offered_depth, self.max_depth)):
            self.offered[d] += 1.0
            if d < accepted_len:
                self.accepted[d] += 1.0

    def rate(self, depth: int) -> float:
        d = min(depth, self.max_depth - 1)
        return self.accepted[d] / self.offered[d]


def plan_tree_shape(stats: AcceptanceStats, budget: int,
                    max_depth: int = 8, max_width: int = 4,
                    cost_ratio: float = 0.0) -> List[int]:
    """Widths per depth maximizing expected accepted length under a total
    node budget. Greedy marginal-gain allocation (the reference's dynamic
    width optimization, :74-165): start with a depth-1 chain and repeatedly
    add the node with the best expected marginal accepted-length gain.

    cost_ratio: draft-step cost in TARGET-step units (e.g. ~0.3 for a
    4-bit self-draft). A node is only added while its marginal expected
    accepted tokens exceed what the same wall time would emit by plain
    decoding — the tokens/sec-optimal stopping rule the reference's
    fixed-budget planner lacks. With a low-acceptance draft the tree
    collapses toward a single probe node instead of burning budget.
    """
    widths = [1]
    nodes = 1

    def gain_extend_depth() -> float:
        d = len(widths)
        if d >= max_depth:
            return -1.0
        p = 1.0
        for i in range(d):
            p *= min(1.0, stats.rate(i) * min(widths[i], max_width) ** 0.5)
        return p * stats.rate(d)

    def gain_widen(d: int) -> float:
        if widths[d] >= max_width:
            return -1.0
        p = 1.0
        for i in range(d):
            p *= min(1.0, stats.rate(i))
        # diminishing branching benefit ~ sqrt growth
        cur = math.sqrt(widths[d])
        return p * stats.rate(d) * (math.sqrt(widths[d] + 1) - cur)

    while nodes < budget:
        options = [("deep", gain_extend_depth())]
        options += [(("wide", d), gain_widen(d)) for d in range(len(widths))]
        kind, g = max(options, key=lambda kv: kv[1])
        if g <= cost_ratio:
            break
        if kind == "deep":
            widths.append(1)
        else:
            widths[kind[1]] += 1
        nodes += 1
    return widths
