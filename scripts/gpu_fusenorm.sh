#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== fused-norm parity tests ==="
timeout 600 python -m pytest tests/test_gpu_kernels.py -m gpu -q -p no:cacheprovider \
  -k "fused_norm or skinny or quantized or block" > gpurun_out/fusenorm_tests.log 2>&1
echo "tests rc=$?"; tail -4 gpurun_out/fusenorm_tests.log
echo "=== bench A/B: fused vs separate norms ==="
BBAMD_FUSE_NORM=0 timeout 420 python bench.py --gpus 1 --steps 48 --warmup 12 \
  > gpurun_out/bench_nofuse.json 2> gpurun_out/bench_nofuse.err
echo "nofuse rc=$?"; cat gpurun_out/bench_nofuse.json
timeout 420 python bench.py --gpus 1 --steps 48 --warmup 12 \
  > gpurun_out/bench_fuse.json 2> gpurun_out/bench_fuse.err
echo "fuse rc=$?"; cat gpurun_out/bench_fuse.json
echo DONE
