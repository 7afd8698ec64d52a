"""Tree verification (parity: reference spec_decoding_verify.py:44-154 —
greedy path acceptance and SpecInfer-style stochastic edge acceptance)."""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from bloombee_amd.spec.tree import TokenTree


def verify_tree_greedy(tree: TokenTree, logits: torch.Tensor,
                       prefix_logits: torch.Tensor,
                       start: Optional[int] = None,
                       ) -> Tuple[List[int], int]:
    """Greedy target verification.

    logits: (T, V) — target logits AT each tree node (i.e. the distribution
    for the token FOLLOWING node i). prefix_logits: (V,) — target logits at
    the last committed token (distribution for the first tree level).

    Returns (accepted_path linear indices, bonus_token):
    walk from the prefix: at each step the target's argmax must equal one of
    the current children's tokens; the matching child is accepted and the
    walk continues from it. When no child matches, the argmax token itself is
    the bonus token (ref verify_path :58-106).
    """
    accepted: List[int] = []
    cur_children = tree.children(start) if start is not None else tree.roots()
    cur_logits = prefix_logits
    while True:
        want = int(cur_logits.argmax(-1))
        match = next((c for c in cur_children if tree.tokens[c] == want), None)
        if match is None:
            return accepted, want
        accepted.append(match)
        cur_logits = logits[match]
        cur_children = tree.children(match)


def verify_tree_sampling(tree: TokenTree, logits: torch.Tensor,
                         prefix_logits: torch.Tensor,
                         generator: Optional[torch.Generator] = None,
                         temperature: float = 1.0,
                         start: Optional[int] = None,
                         ) -> Tuple[List[int], int]:
    """SpecInfer-style stochastic verification (ref verify_edge :108-154).

    At each node, children are tried in order: child c with draft prob q_c
    and target prob p_c is accepted with min(1, p_c / q_c); on rejection the
    target residual is the reference's residual_distribution — when the tree
    carries full draft distributions (tree.dists, populated by the drafter)
    the residual is normalize(clamp(p_target - p_draft, 0)), which makes the
    output distribution-equivalent to sampling the target directly. Trees
    without stored distributions fall back to subtracting only the rejected
    token's scalar draft prob (p[tok] <- max(p[tok] - q_c, 0)): a biased
    approximation that over-weights tokens the draft liked but didn't pick.
    """
    accepted: List[int] = []
    cur_children = tree.children(start) if start is not None else tree.roots()
    cur_logits = prefix_logits
    while True:
        p = torch.softmax(cur_logits.float() / max(temperature, 1e-6), -1)
        matched = None
        for c in cur_children:
            tok = tree.tokens[c]
            q = max(tree.probs[c], 1e-9)
            r = torch.rand((), generator=generator).item()
            if r < min(1.0, float(p[tok]) / q):
                matched = c
                break
            # reject: renormalize the residual (full-dist subtraction when
            # the drafter recorded q(.|ctx); scalar fallback otherwise)
            q_dist = tree.dists[c] if c < len(tree.dists) else None
            if q_dist is not None:
                p = torch.clamp(p - q_dist.float().to(p.device), min=0.0)
            else:
                p[tok] = torch.clamp(p[tok] - q, min=0.0)
            s = p.sum()
            if s <= 0:
                p = torch.ones_like(p) / p.numel()
            else:
                p = p / s
        if matched is None:
            bonus = int(torch.multinomial(p, 1, generator=generator))
            return accepted, bonus
        accepted.append(matched)
        cur_logits = logits[matched]
        cur_children = tree.children(matched)
