"""Batch-sliced view of a SessionHandle for micro-batch pipelining.

The reference slices every request tensor into micro-batch views and builds
per-block InferenceMetadata with batch offsets (server/microbatch.py:27-60,
138). Paged-first, the same thing is a row slice of the page table — the
kernels' batch index b maps to page_table row b, so a contiguous row window
is a zero-copy view.
"""
from __future__ import annotations

import torch

from bloombee_amd.kv.paged import SessionHandle


class SessionView:
    def __init__(self, handle: SessionHandle, b0: int, b1: int):
        self.handle = handle
        self.b0, self.b1 = b0, b1
        self.batch_size = b1 - b0

    @property
    def seqs(self):
        return self.handle.seqs[self.b0:self.b1]

    def page_table(self) -> torch.Tensor:
        return self.handle.page_table()[self.b0:self.b1]

    def k_pages(self, layer: int) -> torch.Tensor:
        return self.handle.k_pages(layer)

    def v_pages(self, layer: int) -> torch.Tensor:
        return self.handle.v_pages(layer)

    @property
    def pos_offset(self) -> int:
        return self.handle.pos_offset

    def host_prefix(self, layer: int):
        hp = self.handle.host_prefix(layer)
        if hp is None:
            return None
        return hp[0][self.b0:self.b1], hp[1][self.b0:self.b1]
