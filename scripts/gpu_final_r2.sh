#!/bin/bash
# Round-2 final GPU validation batch (run via gpurun).
set -x
mkdir -p gpurun_out
cd /root/repo

echo "=== 1. full GPU test suite ==="
timeout 900 python -m pytest tests -m gpu -q -p no:cacheprovider \
  > gpurun_out/final_pytest_gpu.log 2>&1
echo "pytest rc=$?"; tail -4 gpurun_out/final_pytest_gpu.log

echo "=== 2. tree-verify kernel tests ==="
timeout 300 python -m pytest tests/test_treekernel.py -m treekernel -q -p no:cacheprovider \
  > gpurun_out/final_treekernel.log 2>&1
echo "treekernel rc=$?"; tail -3 gpurun_out/final_treekernel.log

echo "=== 3. ctx-8k pipeline bench (prompt 8000) ==="
timeout 420 python bench.py --gpus 1 --steps 16 --warmup 4 --prompt 8000 \
  > gpurun_out/final_bench_ctx8k.json 2> gpurun_out/final_bench_ctx8k.err
echo "ctx8k rc=$?"; cat gpurun_out/final_bench_ctx8k.json

echo "=== 4. fresh flagship pipeline bench ==="
timeout 420 python bench.py --gpus 1 --steps 64 --warmup 16 \
  > gpurun_out/final_bench_pipeline.json 2> gpurun_out/final_bench_pipeline.err
echo "pipeline rc=$?"; cat gpurun_out/final_bench_pipeline.json

echo "=== 5. fresh swarm-mode bench (serving stack) ==="
timeout 420 python bench.py --gpus 1 --steps 64 --warmup 16 --mode swarm \
  > gpurun_out/final_bench_swarm.json 2> gpurun_out/final_bench_swarm.err
echo "swarm rc=$?"; cat gpurun_out/final_bench_swarm.json

echo "=== 6. rocprofv3 kernel stats of the swarm-mode bench (ROUND3 item 11) ==="
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_swarm -o swarm \
  -- python bench.py --gpus 1 --steps 12 --warmup 4 --mode swarm \
  > gpurun_out/final_prof_swarm.log 2>&1
echo "rocprof rc=$?"
find gpurun_out/prof_swarm -name '*stats*' | head -5
# keep only the stats csv (dbs can be large)
find gpurun_out/prof_swarm -name '*.db' -delete 2>/dev/null

echo "=== 7. qwen3 W4 routing sanity on GPU ==="
timeout 300 python - > gpurun_out/final_qwen3_w4.log 2>&1 <<'PYEOF'
import torch, time
from bloombee_amd.engine import LocalEngine
torch.manual_seed(0)
eng = LocalEngine("qwen3-0.6b", device="cuda:0", quantize_q4=True)
blk = eng.stack.blocks[0]
assert getattr(blk, "_w4", None), "W4 table missing on qwen3 block"
assert blk.qkv_w.numel() == 0, "bf16 weights not dropped"
ids = torch.randint(0, 1000, (2, 16), device="cuda:0")
out = eng.generate_greedy(ids, 8)
torch.cuda.synchronize()
print("qwen3 W4 OK, out shape", tuple(out.shape))
PYEOF
echo "qwen3w4 rc=$?"; cat gpurun_out/final_qwen3_w4.log
du -sh gpurun_out/ | tail -1
echo ALL DONE
