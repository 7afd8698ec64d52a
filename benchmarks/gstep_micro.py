import sys, time, torch
sys.path.insert(0, "/root/repo")
from bloombee_amd.engine import LocalEngine
eng = LocalEngine("llama-spec-draft", device="cuda:0", seed=1, kv_max_tokens=1<<13)
kv = eng.kv_pool.allocate(1, 4096)
step = eng.make_graphed_decoder(kv)
for j in range(16):
    kv.extend(1); r = step(5, j)
torch.cuda.synchronize()

def bench(name, fn, n=200):
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for j in range(n):
        fn(j)
    torch.cuda.synchronize()
    print(f"{name}: {(time.monotonic()-t0)/n*1e6:.1f} us/node")
    kv.rollback()

bench("full graphed node", lambda j: (kv.extend(1, speculative=True), step(7, 16+j)))
# eager for reference
def f4(j):
    kv.extend(1, speculative=True)
    pos = kv.seqs[0].l_spec - 1
    h = eng._embed(torch.tensor([[7]]))
    sp = torch.tensor([pos], dtype=torch.int32, device=eng.device)
    h = eng.stack.forward_inference(h, kv, sp)
    lg = eng.logits_for(h[:, -1]).float()[0]
    p = torch.softmax(lg, -1)
    t = int(p.argmax())
bench("eager node", f4)
