import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bloombee_amd.engine import LocalEngine
import bloombee_amd.ops.interface as iface

case = sys.argv[1]
if case.endswith("_noskinny"):
    _fl = torch.nn.functional.linear
    def plain(x, w, residual=None, bias=None):
        y = _fl(x, w, bias)
        if residual is not None:
            y = y + residual.view_as(y)
        return y
    iface.linear = plain
    import bloombee_amd.ops as O
    O.linear = plain
    case = case[:-len("_noskinny")]

B, T = {"b1t16": (1,16), "b2t33": (2,33), "b1t33": (1,33)}[case]
eng = LocalEngine("llama-mini-gpu", device="cuda:0", seed=3, kv_max_tokens=1<<14)
ids = torch.randint(0, 1000, (B, T), generator=torch.Generator().manual_seed(1))
out = eng.generate_greedy(ids, 4)
torch.cuda.synchronize()
print(case, "OK", flush=True)
