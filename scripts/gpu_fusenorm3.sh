#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== parity + ss-chain tests (fuse enabled) ==="
BBAMD_FUSE_NORM=1 timeout 600 python -m pytest tests/test_gpu_kernels.py -m gpu -q -p no:cacheprovider \
  -k "fused_norm or ss_chain or skinny or block" > gpurun_out/fusenorm3_tests.log 2>&1
echo "tests rc=$?"; tail -4 gpurun_out/fusenorm3_tests.log
echo "=== micro: mode2 (folded) vs plain ==="
timeout 300 python - > gpurun_out/fusenorm3_micro.log 2>&1 <<'PYEOF'
import torch
from bloombee_amd import ops
DEV = "cuda:0"
def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True); e = torch.cuda.Event(True); s.record()
    for _ in range(iters): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e-3
for M, N, K, tag in [(32, 6144, 4096, "qkv"), (32, 4096, 4096, "o"),
                     (32, 28672, 4096, "gate_up"), (32, 4096, 14336, "down")]:
    x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV)
    ssin = torch.rand(K // 64 * 32, device=DEV) + 1.0
    ssout = torch.empty(N // 64 * 32, device=DEV)
    r = torch.randn(M, N, dtype=torch.bfloat16, device=DEV)
    t0 = timeit(lambda: ops.hip_ops.gemm_skinny(x, w, None, None, 0))
    t1 = timeit(lambda: ops.hip_ops.gemm_skinny(x, w, None, None, 0, None, 1e-5, ssin, None, 2))
    t2 = timeit(lambda: ops.hip_ops.gemm_skinny(x, w, r, None, 0, None, 0.0, None, ssout, 0))
    t3 = timeit(lambda: ops.hip_ops.gemm_skinny(x, w, None, None, 0, None, 1e-5, ssin, ssout, 2))
    print(f"{tag} {M}x{N}x{K}: plain {t0*1e6:6.1f}  m2norm {t1*1e6:6.1f} "
          f"(+{(t1-t0)*1e6:4.1f})  ssout {t2*1e6:6.1f} (+{(t2-t0)*1e6:4.1f})  "
          f"both {t3*1e6:6.1f}")
PYEOF
cat gpurun_out/fusenorm3_micro.log
echo "=== bench A/B round 3 ==="
BBAMD_FUSE_NORM=0 timeout 420 python bench.py --gpus 1 --steps 48 --warmup 12 \
  > gpurun_out/bench_nofuse3.json 2>&1
grep -o '"value": [0-9.]*' gpurun_out/bench_nofuse3.json
BBAMD_FUSE_NORM=1 timeout 420 python bench.py --gpus 1 --steps 48 --warmup 12 \
  > gpurun_out/bench_fuse3.json 2>&1
grep -o '"value": [0-9.]*' gpurun_out/bench_fuse3.json
echo DONE
