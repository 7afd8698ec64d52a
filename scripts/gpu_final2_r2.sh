#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== spec depth 20/24 (new kernels) ==="
for D in 20 24; do
  timeout 420 python benchmarks/spec_trained.py --max-depth $D --node-budget $((D+2)) \
    > gpurun_out/spec_d${D}.log 2>&1
  echo "d$D rc=$?"; tail -1 gpurun_out/spec_d${D}.log
done
echo "=== full GPU suite (committed state) ==="
timeout 800 python -m pytest tests -m gpu -q -p no:cacheprovider > gpurun_out/final2_pytest.log 2>&1
echo "pytest rc=$?"; tail -2 gpurun_out/final2_pytest.log
echo "=== smoke ==="
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke(); print('SMOKE OK')" 2>&1 | tail -1
echo "=== record benches ==="
timeout 420 python bench.py --gpus 1 --steps 64 --warmup 16 > gpurun_out/final2_pipe.json 2>&1
grep -o '"value": [0-9.]*' gpurun_out/final2_pipe.json
timeout 420 python bench.py --gpus 1 --steps 64 --warmup 16 --mode swarm > gpurun_out/final2_swarm.json 2>&1
grep -o '"value": [0-9.]*' gpurun_out/final2_swarm.json
echo DONE
