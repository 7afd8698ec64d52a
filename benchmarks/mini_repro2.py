import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bloombee_amd.engine import LocalEngine

case = sys.argv[1]
DEV = "cuda:0"
if case == "greedy8":
    eng = LocalEngine("llama-mini-gpu", device=DEV, seed=3, kv_max_tokens=1<<14)
    ids = torch.randint(0, 1000, (1, 33), generator=torch.Generator().manual_seed(1))
    out = eng.generate_greedy(ids, 8)
elif case == "specsteps":
    eng = LocalEngine("llama-mini-gpu", device=DEV, seed=3, kv_max_tokens=1<<14)
    ids = torch.randint(0, 1000, (1, 33), generator=torch.Generator().manual_seed(1))
    kv = eng.kv_pool.allocate(1, 41)
    eng.prefill(ids, kv)
    for i in range(4):
        kv.extend(1, speculative=True)
        h = eng._embed(torch.tensor([[5]]))
        sp = torch.tensor([33 + i], dtype=torch.int32)
        eng.stack.forward_inference(h, kv, sp)
    kv.rollback(); kv.close()
elif case == "with8b":
    tgt = LocalEngine("llama-3-8b", device=DEV, seed=0, kv_max_tokens=1<<15)
    eng = LocalEngine("llama-mini-gpu", device=DEV, seed=3, kv_max_tokens=1<<14)
    ids = torch.randint(0, 1000, (1, 33), generator=torch.Generator().manual_seed(1))
    out = eng.generate_greedy(ids, 4)
elif case == "tree8b":
    tgt = LocalEngine("llama-3-8b", device=DEV, seed=0, kv_max_tokens=1<<15)
    ids = torch.randint(0, 1000, (1, 32), generator=torch.Generator().manual_seed(1))
    kv = tgt.kv_pool.allocate(1, 512)
    tok = tgt.prefill(ids, kv)
    from bloombee_amd.spec.tree import TokenTree
    tree = TokenTree(); tree.add(int(tok), -1, 1.0)
    for t in range(5):
        tree.add(100 + t, t, 0.5)
    prefix = kv.seqs[0].l_acc
    toks = tree.token_tensor().view(1, -1).to(DEV)
    pos = tree.position_ids(prefix).view(1, -1)
    mask = tree.attention_mask().unsqueeze(0)
    kv.extend(len(tree), speculative=True)
    hid = tgt._embed(toks)
    sp = torch.full((1,), prefix, dtype=torch.int32, device=DEV)
    h = tgt.stack.forward_inference(hid, kv, sp, pos.int().to(DEV), tree_mask=mask.to(DEV))
    kv.reorder_and_commit([[0, 1]])
torch.cuda.synchronize()
print(case, "OK", flush=True)
