#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== spec_trained depth sweep ==="
for D in 8 12 16; do
  timeout 420 python benchmarks/spec_trained.py --max-depth $D --node-budget $((D+2)) \
    > gpurun_out/spec_depth${D}.log 2>&1
  echo "depth $D rc=$?"; grep -E "tokens/s|accepted|speedup" gpurun_out/spec_depth${D}.log | tail -6
done
echo "=== swarm-mode bench at B64 ==="
timeout 420 python bench.py --gpus 1 --steps 32 --warmup 8 --batch-per-gpu 64 --mode swarm \
  > gpurun_out/bench_swarm_b64.json 2> gpurun_out/bench_swarm_b64.err
echo "swarm b64 rc=$?"; cat gpurun_out/bench_swarm_b64.json
echo "=== pipeline bench at B64 (same box, for the overhead ratio) ==="
timeout 420 python bench.py --gpus 1 --steps 32 --warmup 8 --batch-per-gpu 64 \
  > gpurun_out/bench_pipe_b64.json 2> gpurun_out/bench_pipe_b64.err
echo "pipe b64 rc=$?"; cat gpurun_out/bench_pipe_b64.json
echo DONE
