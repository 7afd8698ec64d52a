#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== FULL GPU suite with fusion default ON ==="
timeout 900 python -m pytest tests -m gpu -q -p no:cacheprovider \
  > gpurun_out/fusenorm4_full.log 2>&1
echo "full rc=$?"; tail -3 gpurun_out/fusenorm4_full.log
echo "=== swarm bench with fusion ==="
timeout 420 python bench.py --gpus 1 --steps 48 --warmup 12 --mode swarm \
  > gpurun_out/bench_swarm_fuse.json 2>&1
grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/bench_swarm_fuse.json
echo "=== pipeline bench (committed default) ==="
timeout 420 python bench.py --gpus 1 --steps 64 --warmup 16 \
  > gpurun_out/bench_pipe_fuse.json 2>&1
grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/bench_pipe_fuse.json
echo DONE
