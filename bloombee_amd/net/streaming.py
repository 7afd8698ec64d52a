"""Chunked tensor streaming for large unary payloads.

Parity: the reference splits forward/backward payloads bigger than
MAX_UNARY_PAYLOAD_SIZE // 2 into stream chunks
(client/remote_forward_backward.py:46-118 split_for_streaming); round 1
sent one unbounded frame (up to the 2 GB frame cap) per call. Tensors are
flattened and sent as row-sliced parts; the first part of each tensor
carries shape/dtype, the terminator carries the tensor count.

The device data plane (net/channels.py) makes chunking moot — RCCL moves
the payload out-of-band — so callers pick streaming only for plain-TCP
peers with oversized payloads.
"""
from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch

# ceiling for one unary frame's tensor payload; streams split at half this
MAX_UNARY_PAYLOAD_BYTES = int(os.environ.get("BBAMD_MAX_UNARY_PAYLOAD",
                                             str(32 << 20)))

_DTYPES = {
    "torch.bfloat16": torch.bfloat16, "torch.float16": torch.float16,
    "torch.float32": torch.float32, "torch.float64": torch.float64,
    "torch.int8": torch.int8, "torch.uint8": torch.uint8,
    "torch.int16": torch.int16, "torch.int32": torch.int32,
    "torch.int64": torch.int64, "torch.bool": torch.bool,
}


def payload_nbytes(tensors: List[torch.Tensor]) -> int:
    return sum(t.numel() * t.element_size() for t in tensors)


async def send_tensors_chunked(stream, tensors: List[torch.Tensor],
                               codec: str = "raw",
                               meta: Optional[dict] = None) -> None:
    """Send `tensors` as flattened row-sliced parts followed by a {"eot"}
    terminator (which also carries `meta`)."""
    chunk = max(1, MAX_UNARY_PAYLOAD_BYTES // 2)
    for i, t in enumerate(tensors):
        t = t.detach().contiguous().cpu()
        flat = t.reshape(-1)
        rows = max(1, chunk // max(1, t.element_size()))
        nparts = max(1, (flat.numel() + rows - 1) // rows)
        for j in range(nparts):
            head = {"i": i, "part": j, "parts": nparts}
            if j == 0:
                head["shape"] = list(t.shape)
                head["dtype"] = str(t.dtype)
            await stream.send(head, [flat[j * rows:(j + 1) * rows]],
                              codec=codec)
    await stream.send({"eot": True, "n": len(tensors), **(meta or {})})


async def recv_tensors_chunked(stream) -> Tuple[dict, List[torch.Tensor]]:
    """Collect parts until the {"eot"} terminator; returns (eot meta,
    reassembled tensors). Raises on a truncated stream."""
    shapes: dict = {}
    parts: dict = {}
    while True:
        item = await stream.recv()
        if item is None:
            raise ConnectionError("stream ended before eot terminator")
        m, ts = item
        if m.get("eot"):
            n = int(m["n"])
            out = []
            for i in range(n):
                if i not in shapes:
                    raise ConnectionError(f"missing tensor {i} in stream")
                shape, dtype = shapes[i]
                got = parts[i]
                if len(got) != got[-1][2]:
                    raise ConnectionError(f"missing parts for tensor {i}")
                flat = torch.cat([p for _, p, _ in sorted(got,
                                                          key=lambda x: x[0])])
                out.append(flat.reshape(shape).to(dtype))
            return m, out
        i, j, nparts = int(m["i"]), int(m["part"]), int(m["parts"])
        if j == 0:
            shapes[i] = (m["shape"], _DTYPES[m["dtype"]])
        parts.setdefault(i, []).append((j, ts[0], nparts))
