"""Weight offload: FlexGen-equivalent streaming re-designed for MI355X.

Reference: flexgen_utils/pytorch_backend.py TorchDevice/TorchDisk/TorchLink +
models/llama/block.py per-(i,j,k) load_weight overlap. Re-design (SURVEY.md
§7 step 7): block-granular host tier + a double-buffered HBM arena.

  * Offloaded blocks' parameters live on host (pinned when a GPU is present;
    optionally 4-bit group-quantized, optionally np.memmap disk-backed).
  * A side HIP stream copies block i+1's parameters into arena slot (i+1)%2
    while block i computes (the reference's load_weight/compute overlap);
    compressed tensors dequantize on-GPU (quant4.hip) after the H2D copy, so
    the PCIe/xGMI host link moves 4x fewer bytes.
  * Compute never blocks on the copy engine except at the event fence.

288 GB HBM3E sizing note: a llama-2-70B shard of 10 blocks is ~17 GB — the
arena is 2 blocks (~3.5 GB); host DRAM holds the rest.
"""
from __future__ import annotations

import os
import tempfile
from typing import Dict, List, Optional

import numpy as np
import torch

from bloombee_amd import ops
from bloombee_amd.engine import BlockStack
from bloombee_amd.kv.paged import SessionHandle
from bloombee_amd.offload.policy import OffloadPolicy
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


class _HostParam:
    """One offloaded parameter on the host tier."""

    def __init__(self, t: torch.Tensor, pin: bool, compress: bool,
                 disk_dir: Optional[str]):
        self.shape = t.shape
        self.dtype = t.dtype
        self.compressed = compress and t.dtype == torch.bfloat16 and \
            t.shape[-1] % 64 == 0 and t.dim() >= 2
        host = t.detach().cpu()
        if self.compressed:
            packed, scale, zero = ops.quant4_pack(host.reshape(-1, t.shape[-1]))
            self.packed, self.scale, self.zero = packed, scale, zero
            if pin and torch.cuda.is_available():
                self.packed = self.packed.pin_memory()
                self.scale = self.scale.pin_memory()
                self.zero = self.zero.pin_memory()
            self.host = None
        else:
            if disk_dir is not None:
                path = os.path.join(disk_dir, f"w{id(self)}.npy")
                arr = host.view(torch.uint16).numpy() if t.dtype == torch.bfloat16 \
                    else host.numpy()
                m = np.lib.format.open_memmap(path, mode="w+", dtype=arr.dtype,
                                              shape=arr.shape)
                m[:] = arr
                m.flush()
                host = torch.from_numpy(np.lib.format.open_memmap(path, mode="c"))
                if t.dtype == torch.bfloat16:
                    host = host.view(torch.bfloat16)
            elif pin and torch.cuda.is_available():
                host = host.pin_memory()
            self.host = host

    def copy_into(self, dst: torch.Tensor) -> None:
        """Fill `dst` (device arena view) — called inside the copy stream."""
        if self.compressed:
            dev = dst.device
            packed = self.packed.to(dev, non_blocking=True)
            scale = self.scale.to(dev, non_blocking=True)
            zero = self.zero.to(dev, non_blocking=True)
            w = ops.quant4_unpack(packed, scale, zero, dtype=self.dtype)
            dst.copy_(w.view(self.shape))
        else:
            dst.copy_(self.host, non_blocking=True)


class OffloadedBlockStack:
    """Wraps a BlockStack: keeps the first `resident` blocks in HBM, streams
    the rest block-by-block through a double-buffered arena."""

    def __init__(self, stack: BlockStack, policy: OffloadPolicy,
                 disk_dir: Optional[str] = None):
        self.stack = stack
        self.policy = policy
        self.config = stack.config
        n = len(stack.blocks)
        self.resident = min(n, int(round(n * policy.weight_gpu_percent / 100.0)))
        self.device = stack.device
        self.on_gpu = self.device.type == "cuda"
        self._copy_stream = (torch.cuda.Stream(self.device)
                             if self.on_gpu else None)
        self._events: Dict[int, torch.cuda.Event] = {}
        self._done_events: Dict[int, torch.cuda.Event] = {}  # slot -> event
        self._arena: List[Dict[str, torch.Tensor]] = [{}, {}]
        self._arena_owner = [-1, -1]
        self._host: Dict[int, Dict[str, _HostParam]] = {}

        if self.resident < n:
            # host copies + free the device params of offloaded blocks
            ndisk = int(round((n - self.resident)
                              * policy.weight_disk_percent / 100.0))
            for i in range(self.resident, n):
                blk = stack.blocks[i]
                use_disk = disk_dir if (i >= n - ndisk) else None
                params = {}
                for name, p in blk.named_parameters():
                    params[name] = _HostParam(p.data, policy.pin_weight,
                                              policy.compress_weight, use_disk)
                self._host[i] = params
            # arena buffers shaped like block[resident] (uniform block shapes)
            proto = stack.blocks[self.resident]
            for slot in range(2):
                self._arena[slot] = {
                    name: torch.empty_like(p.data)
                    for name, p in proto.named_parameters()}
            # drop offloaded device copies
            for i in range(self.resident, n):
                for name, p in stack.blocks[i].named_parameters():
                    p.data = self._arena[0][name]  # placeholder binding
            logger.info("offload: %d/%d blocks resident, %d on disk, "
                        "compress=%s", self.resident, n, ndisk,
                        policy.compress_weight)

    # ------------------------------------------------------------------
    def _fill(self, i: int, slot: int) -> None:
        arena = self._arena[slot]
        if self.on_gpu:
            with torch.cuda.stream(self._copy_stream):
                # don't overwrite a slot whose previous block may still have
                # kernels in flight on the compute stream
                done = self._done_events.pop(slot, None)
                if done is not None:
                    self._copy_stream.wait_event(done)
                for name, hp in self._host[i].items():
                    hp.copy_into(arena[name])
                    # arena tensors were allocated on the default stream; tell
                    # the caching allocator they are used on the copy stream
                    arena[name].record_stream(self._copy_stream)
                ev = torch.cuda.Event()
                ev.record(self._copy_stream)
                self._events[i] = ev
        else:
            for name, hp in self._host[i].items():
                hp.copy_into(arena[name])
        self._arena_owner[slot] = i

    def _bind(self, i: int, slot: int) -> None:
        blk = self.stack.blocks[i]
        arena = self._arena[slot]
        for name, p in blk.named_parameters():
            p.data = arena[name]

    def _ensure(self, i: int) -> None:
        slot = i % 2
        if self._arena_owner[slot] != i:
            self._fill(i, slot)
        if self.on_gpu and i in self._events:
            torch.cuda.current_stream().wait_event(self._events.pop(i))
        self._bind(i, slot)

    @torch.no_grad()
    def forward_inference(self, hidden: torch.Tensor, kv: SessionHandle,
                          start_pos: torch.Tensor,
                          position_ids=None, tree_mask=None,
                          deep_prompts=None) -> torch.Tensor:
        n = len(self.stack.blocks)
        sp0 = int(start_pos[0]) if deep_prompts is not None else 0
        pre = deep_prompts.shape[1] if deep_prompts is not None else 0
        n_over = min(pre - sp0, hidden.shape[1]) if sp0 < pre else 0
        # prefetch the first offloaded block before the resident prefix runs
        if self.resident < n and self.policy.overlap:
            if self._arena_owner[self.resident % 2] != self.resident:
                self._fill(self.resident, self.resident % 2)
        for i, blk in enumerate(self.stack.blocks):
            if i >= self.resident:
                self._ensure(i)
                if self.policy.overlap and i + 1 < n:
                    nxt = i + 1
                    if self._arena_owner[nxt % 2] != nxt:
                        self._fill(nxt, nxt % 2)
            if n_over > 0:
                hidden[:, :n_over] += deep_prompts[i, sp0:sp0 + n_over].to(hidden.dtype)
            if tree_mask is not None:
                hidden = blk.forward_inference(hidden, kv, start_pos,
                                               position_ids, tree_mask=tree_mask)
            else:
                hidden = blk.forward_inference(hidden, kv, start_pos,
                                               position_ids)
            if i >= self.resident and self.on_gpu:
                ev = torch.cuda.Event()
                ev.record(torch.cuda.current_stream())
                self._done_events[i % 2] = ev
        return hidden

    def forward_train(self, hidden: torch.Tensor, start_pos: int = 0,
                      deep_prompts=None):
        pre = deep_prompts.shape[1] if deep_prompts is not None else 0
        for i, blk in enumerate(self.stack.blocks):
            if i >= self.resident:
                self._ensure(i)
            if pre > 0 and hidden.shape[1] >= pre:
                dp = deep_prompts[i].to(hidden.dtype).unsqueeze(0)
                hidden = torch.cat([hidden[:, :pre] + dp, hidden[:, pre:]], 1)
            hidden = blk.forward_train(hidden, start_pos)
        return hidden

    # passthroughs
    def make_kv(self, max_tokens: int):
        return self.stack.make_kv(max_tokens)

    @property
    def blocks(self):
        return self.stack.blocks
