import sys; sys.path.insert(0, "/root/repo")
import torch
from bloombee_amd.engine import BlockStack
from bloombee_amd.models.base import resolve_config

mode = sys.argv[1]
DEV = "cuda:0"
cfg = resolve_config("llama-2-70b")
nb = int(sys.argv[2]) if len(sys.argv) > 2 else 4
stack = BlockStack(cfg, 0, nb, device=DEV, seed=0)
print("stack built", flush=True)
if mode == "offload":
    from bloombee_amd.offload import OffloadPolicy, OffloadedBlockStack
    stack = OffloadedBlockStack(stack, OffloadPolicy(weight_gpu_percent=50.0))
    print("offloaded", flush=True)
kv = stack.make_kv(1 << 15)
B, T = 8, 128
h = kv.allocate(B, T + 20)
hid = (torch.randn(B, T, cfg.hidden_size) * 0.02).to(cfg.dtype).to(DEV)
h.extend(T)
out = stack.forward_inference(hid, h, torch.zeros(B, dtype=torch.int32, device=DEV))
torch.cuda.synchronize()
print("prefill ok", flush=True)
one = (torch.randn(B, 1, cfg.hidden_size) * 0.02).to(cfg.dtype).to(DEV)
for i in range(8):
    sp = torch.full((B,), T + i, dtype=torch.int32, device=DEV)
    h.extend(1)
    stack.forward_inference(one, h, sp)
    torch.cuda.synchronize()
    print("step", i, flush=True)
h.close()
print("DONE", flush=True)
