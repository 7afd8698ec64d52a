#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 800 python -m pytest tests -m gpu -q -p no:cacheprovider > gpurun_out/wrap_pytest.log 2>&1
echo "gpu suite rc=$?"; tail -1 gpurun_out/wrap_pytest.log
timeout 200 python -m pytest tests/test_treekernel.py -m treekernel -q -p no:cacheprovider 2>&1 | tail -1
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke(); print('SMOKE OK')" 2>&1 | tail -1
timeout 420 python bench.py --gpus 1 --steps 96 --warmup 16 > gpurun_out/wrap_pipe.json 2>&1
grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/wrap_pipe.json
timeout 420 python bench.py --gpus 1 --steps 64 --warmup 16 --mode swarm > gpurun_out/wrap_swarm.json 2>&1
grep -o '"value": [0-9.]*' gpurun_out/wrap_swarm.json
timeout 420 python bench.py --gpus 1 --steps 16 --warmup 4 --prompt 8000 > gpurun_out/wrap_ctx8k.json 2>&1
grep -o '"value": [0-9.]*' gpurun_out/wrap_ctx8k.json
echo WRAP DONE
