#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== swarm-mode kernel stats (csv this time) ==="
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv \
  -d gpurun_out/prof_swarm2 -o swarm \
  -- python bench.py --gpus 1 --steps 12 --warmup 4 --mode swarm \
  > gpurun_out/final_prof_swarm2.log 2>&1
echo "swarmprof rc=$?"
find gpurun_out/prof_swarm2 -name '*.db' -delete 2>/dev/null
find gpurun_out/prof_swarm2 -name '*stats*'
head -12 $(find gpurun_out/prof_swarm2 -name '*kernel_stats*' | head -1) 2>/dev/null
bash scripts/gpu_pmc_utcl2.sh
