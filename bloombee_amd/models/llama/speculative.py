"""Distributed speculative generation (parity: reference
models/llama/speculative_model.py DistributedLlamaForSpeculativeGeneration
:29-968 — draft trees from a small local model, one tree-verify forward per
round over the swarm, paged commit/rollback of accepted nodes).

Greedy invariant: with greedy verification the emitted tokens are EXACTLY
the plain greedy decode of the target model (tested against LocalEngine).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from bloombee_amd.engine import LocalEngine
from bloombee_amd.models.llama.model import DistributedLlamaForCausalLM
from bloombee_amd.spec.drafter import MultiDrafter
from bloombee_amd.spec.tree import TokenTree
from bloombee_amd.spec.verify import verify_tree_greedy, verify_tree_sampling
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


class _RestrictedTree:
    """Tree view whose children() hides pruned nodes (they can never be
    accepted); tokens/probs/roots pass through."""

    def __init__(self, tree: TokenTree, allowed):
        self._t = tree
        self._allowed = allowed
        self.tokens = tree.tokens
        self.probs = tree.probs
        self.parents = tree.parents

    def roots(self):
        return [r for r in self._t.roots() if r in self._allowed]

    def children(self, i):
        return [c for c in self._t.children(i) if c in self._allowed]


def _restrict(tree: TokenTree, allowed) -> "_RestrictedTree":
    return _RestrictedTree(tree, allowed)


class DistributedLlamaForSpeculativeGeneration(DistributedLlamaForCausalLM):
    """Adds `generate_speculative` on top of the distributed causal LM.

    draft_model: a small LocalEngine (e.g. qwen3-0.6b per BASELINE config 5)
    running ON the client.
    """

    def set_drafter(self, draft_model: LocalEngine, node_budget: int = 8,
                    max_depth: int = 5, n_workers: int = 2):
        self.drafter = MultiDrafter(draft_model, n_workers=n_workers,
                                    node_budget=node_budget, max_depth=max_depth)
        return self

    @torch.no_grad()
    def generate_speculative(self, input_ids: torch.Tensor,
                             max_new_tokens: int = 20,
                             do_sample: bool = False,
                             session=None) -> torch.Tensor:
        assert input_ids.shape[0] == 1, "speculative path is per-sequence"
        assert hasattr(self, "drafter"), "call set_drafter() first"
        B, T = input_ids.shape
        budget = self.drafter.node_budget + self.drafter.max_depth + 2
        own = session is None
        if own:
            session = self.remote.inference_session(
                T + max_new_tokens + 1 + budget * 2)
            session.allow_push = False

        history = input_ids[0].tolist()
        out: List[int] = []
        try:
            # committed prefill; its last logits decide the pending token
            hidden = self.embed(input_ids)
            hidden = session.step(hidden)
            prev_logits = self.lm_head(self.final_norm(hidden[:, -1:]))[0, -1].float()
            pending = int(prev_logits.argmax(-1))

            while len(out) < max_new_tokens:
                # ---- draft a tree under the pending token ----
                draft_hist = torch.tensor(history + [pending])
                sub = self.drafter.build_tree(draft_hist)
                tree = TokenTree()
                tree.add(pending, -1, 1.0)          # node 0: forced pending
                for i in range(len(sub)):
                    parent = 0 if sub.parents[i] == -1 else sub.parents[i] + 1
                    tree.add(sub.tokens[i], parent, sub.probs[i])

                # ---- one tree-verify forward over the swarm ----
                prefix_len = session.position
                toks = tree.token_tensor().view(1, -1)
                pos_ids = tree.position_ids(prefix_len).view(1, -1)
                mask = tree.attention_mask().unsqueeze(0)
                h = self.embed(toks)
                h, keep = session.spec_step(
                    h, pos_ids.int(), mask,
                    tree={"tokens": tree.tokens, "parents": tree.parents})
                logits_k = self.lm_head(self.final_norm(h))[0].float()
                if keep is not None:
                    # last-block pruner flattened to kept rows: scatter back;
                    # pruned nodes can never be accepted (restricted walk,
                    # ref _restore_hidden_states, inference_session.py:696-800)
                    T = len(tree)
                    V = logits_k.shape[-1]
                    logits = torch.zeros(T, V)
                    logits[keep] = logits_k
                    allowed = set(keep)
                    pruned_tree = _restrict(tree, allowed)
                else:
                    logits = logits_k
                    pruned_tree = tree

                # ---- accept ----
                # node 0 (pending) is committed by construction; verify walks
                # its subtree
                sub_accepted, bonus = (
                    verify_tree_greedy(pruned_tree, logits, logits[0], start=0)
                    if not do_sample else
                    verify_tree_sampling(pruned_tree, logits, logits[0],
                                         start=0))
                accepted = [0] + sub_accepted
                session.spec_commit([accepted])

                emitted = [tree.tokens[i] for i in accepted]
                out.extend(emitted)
                history.extend(emitted)
                self.drafter.record_result(len(sub_accepted),
                                           offered_depth=self.drafter.max_depth)
                pending = bonus
                if len(out) < max_new_tokens:
                    # bonus rides as the next round's node 0
                    pass
            return torch.tensor(out[:max_new_tokens]).view(1, -1)
        finally:
            if own:
                session.close()
