"""Wire codec throughput: Python vs native C++ vs native multithreaded
(ops/hip/wire.h). CPU-only; run anywhere.

    python benchmarks/bench_wire.py [MB]
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from bloombee_amd.net import tensors as T  # noqa: E402


def main():
    mb = int(sys.argv[1]) if len(sys.argv) > 1 else 16
    n = mb * (1 << 20) // 2
    t = (torch.randn(n) * 0.1).bfloat16()
    rows = []
    for codec in ("bsplit+zlib", "bsplit+zlibmt"):
        h, p = T.serialize_tensor(t, codec=codec)
        assert torch.equal(T.deserialize_tensor(h, p), t)
        t0 = time.monotonic()
        for _ in range(5):
            T.serialize_tensor(t, codec=codec)
        enc = (time.monotonic() - t0) / 5
        t0 = time.monotonic()
        for _ in range(5):
            T.deserialize_tensor(h, p)
        dec = (time.monotonic() - t0) / 5
        rows.append({"codec": codec, "payload_mb": mb,
                     "enc_MBps": round(n * 2 / enc / 1e6, 1),
                     "dec_MBps": round(n * 2 / dec / 1e6, 1),
                     "ratio": round(len(p) / (n * 2), 3),
                     "native": T._native is not None})
    for r in rows:
        print(json.dumps(r))


if __name__ == "__main__":
    main()
