import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from bloombee_amd.engine import LocalEngine
from bloombee_amd.spec.drafter import MultiDrafter
from bloombee_amd.spec.tree import TokenTree
from bloombee_amd.spec.verify import verify_tree_greedy

DEV = "cuda:0"
def S(msg):
    torch.cuda.synchronize(); print(msg, flush=True)

tgt = LocalEngine("llama-3-8b", device=DEV, seed=0, kv_max_tokens=1 << 15)
S("target built")
draft = LocalEngine("llama-mini-gpu", device=DEV, seed=3, kv_max_tokens=1 << 14)
S("draft built")
drafter = MultiDrafter(draft, node_budget=8, max_depth=4)
prompt = torch.randint(0, 1000, (1, 32), generator=torch.Generator().manual_seed(1))
kv = tgt.kv_pool.allocate(1, 512)
tok = tgt.prefill(prompt, kv)
S("prefill done")
history = prompt[0].tolist()
pending = int(tok)
for r in range(30):
    sub = drafter.build_tree(torch.tensor(history + [pending]))
    S(f"r{r} drafted {len(sub)}")
    tree = TokenTree(); tree.add(pending, -1, 1.0)
    for i in range(len(sub)):
        tree.add(sub.tokens[i], 0 if sub.parents[i] == -1 else sub.parents[i] + 1, sub.probs[i])
    prefix = kv.seqs[0].l_acc
    toks = tree.token_tensor().view(1, -1).to(DEV)
    pos = tree.position_ids(prefix).view(1, -1)
    mask = tree.attention_mask().unsqueeze(0)
    kv.rollback(); kv.extend(len(tree), speculative=True)
    hid = tgt._embed(toks)
    sp = torch.full((1,), prefix, dtype=torch.int32, device=DEV)
    h = tgt.stack.forward_inference(hid, kv, sp, pos.int().to(DEV), tree_mask=mask.to(DEV))
    S(f"r{r} fwd done prefix={prefix} T={len(tree)}")
    logits = tgt.logits_for(h[0]).float().cpu()
    acc, bonus = verify_tree_greedy(tree, logits, logits[0], start=0)
    accepted = [0] + acc
    kv.reorder_and_commit([accepted])
    S(f"r{r} committed {len(accepted)}")
    emit = [tree.tokens[i] for i in accepted]
    history += emit
    pending = bonus
    drafter.record_result(len(acc), offered_depth=4)
print("ALL DONE", flush=True)
