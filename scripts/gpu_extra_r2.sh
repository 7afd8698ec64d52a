#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== W4 8B engine decode (end-to-end quantized serving) ==="
timeout 420 python - > gpurun_out/w4_8b_decode.log 2>&1 <<'PYEOF'
import torch, time
from bloombee_amd.engine import LocalEngine
def bench(eng, B, steps=32, warm=8):
    ids = torch.randint(0, 1000, (B, 16), device="cuda:0")
    eng.generate_greedy(ids, warm)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    eng.generate_greedy(ids, steps)
    torch.cuda.synchronize()
    return B * steps / (time.perf_counter() - t0)
for B in (1, 8, 32):
    bf = LocalEngine("llama-3-8b", device="cuda:0", kv_max_tokens=1<<17)
    r_bf = bench(bf, B)
    del bf; torch.cuda.empty_cache()
    w4 = LocalEngine("llama-3-8b", device="cuda:0", kv_max_tokens=1<<17, quantize_q4=True)
    r_w4 = bench(w4, B)
    del w4; torch.cuda.empty_cache()
    print(f"B{B}: bf16 {r_bf:7.0f} tok/s   W4 {r_w4:7.0f} tok/s   {r_w4/r_bf:4.2f}x")
PYEOF
cat gpurun_out/w4_8b_decode.log | grep "tok/s"
echo "=== spec over the swarm (new kernels) ==="
timeout 500 python benchmarks/spec_trained.py --swarm --max-depth 12 --node-budget 14 \
  > gpurun_out/spec_swarm_final.log 2>&1
echo "rc=$?"; grep -E "tokens_per_s|speedup|swarm" gpurun_out/spec_swarm_final.log | tail -4
echo "=== sustained flagship (256 steps) ==="
timeout 500 python bench.py --gpus 1 --steps 256 --warmup 16 --prompt 1024 > gpurun_out/sustained.json 2>&1
grep -o '"value": [0-9.]*\|"p50_step_ms": [0-9.]*' gpurun_out/sustained.json
echo DONE
