"""Draft token trees (parity: reference models/llama/spe_dec_tree.py:11-384 —
tree structure, linearization, ancestor-matrix attention masks, per-node
absolute positions for rotary)."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Sequence, Tuple

import torch


@dataclass
class TokenTree:
    """A rooted tree of draft tokens. Node 0's parent is the committed prefix
    (parent = -1). Linear order is insertion order (parents precede children,
    enforced on add)."""

    tokens: List[int] = field(default_factory=list)
    parents: List[int] = field(default_factory=list)
    # optional draft probabilities q(token | parent ctx) for stochastic verify
    probs: List[float] = field(default_factory=list)
    # optional FULL draft distributions q(. | parent ctx) per node — needed
    # for the exact SpecInfer rejection residual (p_target - p_draft)+ in
    # verify_tree_sampling; None entries fall back to the scalar approximation
    dists: List[Optional[torch.Tensor]] = field(default_factory=list)

    def add(self, token: int, parent: int, prob: float = 1.0,
            dist: Optional[torch.Tensor] = None) -> int:
        if not (-1 <= parent < len(self.tokens)):
            raise ValueError(f"bad parent {parent}")
        self.tokens.append(int(token))
        self.parents.append(int(parent))
        self.probs.append(float(prob))
        self.dists.append(dist)
        return len(self.tokens) - 1

    def __len__(self) -> int:
        return len(self.tokens)

    def depth(self, i: int) -> int:
        d = 0
        while self.parents[i] != -1:
            i = self.parents[i]
            d += 1
        return d

    def children(self, i: int) -> List[int]:
        return [j for j, p in enumerate(self.parents) if p == i]

    def roots(self) -> List[int]:
        return [j for j, p in enumerate(self.parents) if p == -1]

    # -- linearized tensors ------------------------------------------------
    def token_tensor(self) -> torch.Tensor:
        return torch.tensor(self.tokens, dtype=torch.long)

    def position_ids(self, prefix_len: int) -> torch.Tensor:
        """Absolute rotary position per node: prefix_len + depth (ref tree
        rotary positions, backend.py:944-1047)."""
        return torch.tensor([prefix_len + self.depth(i) for i in range(len(self))],
                            dtype=torch.long)

    def attention_mask(self) -> torch.Tensor:
        """(T, T) bool ancestor-or-self matrix: node i may attend node j iff
        j is an ancestor of i or i itself. O(n * depth) walk (the reference's
        optimized ancestor walk, sim_end_to_end_decode.py:1-70)."""
        T = len(self)
        m = torch.zeros(T, T, dtype=torch.bool)
        for i in range(T):
            j = i
            while j != -1:
                m[i, j] = True
                j = self.parents[j]
        return m

    def path_to(self, i: int) -> List[int]:
        """Linear indices from a root down to node i (inclusive)."""
        path = []
        while i != -1:
            path.append(i)
            i = self.parents[i]
        return list(reversed(path))

    @classmethod
    def chain(cls, tokens: Sequence[int],
              probs: Optional[Sequence[float]] = None) -> "TokenTree":
        t = cls()
        for d, tok in enumerate(tokens):
            t.add(tok, d - 1, 1.0 if probs is None else probs[d])
        return t
