"""Speculative decoding: draft trees, tree-attention verify, commit/rollback.

Parity targets (reference, SURVEY.md §2.4 llama spec-dec stack):
  spe_dec_tree.py            -> spec/tree.py      (tree + linearization + masks)
  spec_decoding_drafter.py   -> spec/drafter.py   (multi-worker SSM drafting)
  spec_decoding_verify.py    -> spec/verify.py    (greedy + SpecInfer verify)
  spec_decoding_tree_shape.py-> spec/shape.py     (Sequoia-style width planning)
  server/speculative_pruner/ -> spec/pruner.py    (mid-network tree pruning)
"""
from bloombee_amd.spec.tree import TokenTree  # noqa: F401
from bloombee_amd.spec.verify import verify_tree_greedy, verify_tree_sampling  # noqa: F401
