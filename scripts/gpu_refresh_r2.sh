#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== swarm bench (new kernels) ==="
timeout 420 python bench.py --gpus 1 --steps 48 --warmup 12 --mode swarm \
  > gpurun_out/bench_swarm_final.json 2>&1
grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/bench_swarm_final.json
echo "=== spec_trained depth 12 (new kernels) ==="
timeout 420 python benchmarks/spec_trained.py --max-depth 12 --node-budget 14 \
  > gpurun_out/spec_final.log 2>&1
tail -1 gpurun_out/spec_final.log
echo "=== kernel budget of the new decode step ==="
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv \
  -d gpurun_out/prof_final -o fin \
  -- python bench.py --gpus 1 --steps 12 --warmup 4 > gpurun_out/prof_final.log 2>&1
echo "prof rc=$?"
find gpurun_out/prof_final -name '*.db' -delete 2>/dev/null
head -14 $(find gpurun_out/prof_final -name '*kernel_stats*' | head -1)
echo DONE
