"""Dark-shipped tree-mask kernel validation — round-2 gate.

Deliberately NOT in test_gpu_kernels.py (whose module-level gpu mark would
put these in the driver's round-end `-m gpu` run): the fused tree-mask path
is compiled and env-gated (BBAMD_TREE_KERNEL=1) but has not yet run on a
real MI355X. Run `pytest tests/test_treekernel.py -m treekernel` on a GPU
box before flipping the default.
"""
import pytest
import torch

from bloombee_amd.ops import reference as ref
from tests.test_gpu_kernels import DEV, _paged_setup


@pytest.mark.treekernel
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
@pytest.mark.parametrize("shape", [
    (2, 8, 2, 40, 12, 128),   # (B, Hq, Hkv, prefix, tree_nodes, D)
    (1, 4, 4, 100, 24, 128),
    (1, 2, 1, 33, 9, 64),
])
def test_tree_mask_kernel_parity(shape, monkeypatch):
    """Dark-shipped fused tree-mask attention (BBAMD_TREE_KERNEL=1) vs the
    torch composition. NOT part of -m gpu: run `-m treekernel` explicitly
    before flipping the default (round-2 gate)."""
    import importlib

    from bloombee_amd.ops import interface as iface
    from bloombee_amd.spec.tree import TokenTree

    B, Hq, Hkv, prefix, Tq, D = shape
    total = prefix + Tq
    kp, vp, pt, q, k, v, start = _paged_setup(B, Hq, Hkv, total, D, seed=4)
    # random tree over the Tq new nodes
    gen = torch.Generator().manual_seed(0)
    tree = TokenTree()
    for i in range(Tq):
        parent = -1 if i == 0 else int(torch.randint(0, i, (1,), generator=gen))
        tree.add(i, parent, 0.5)
    tm = tree.attention_mask().unsqueeze(0).expand(B, -1, -1).contiguous()
    qn = q[:, :, prefix:].contiguous()  # queries = the Tq tree nodes
    qs = torch.full((B,), prefix, dtype=torch.int32)

    want = ref.attn_paged(qn.float(), kp.float(), vp.float(), pt, qs.long(),
                          tree_mask=tm)
    monkeypatch.setattr(iface, "_TREE_KERNEL", True)
    got = iface.attn_paged(qn.to(DEV), kp.to(DEV), vp.to(DEV), pt.to(DEV),
                           qs.to(DEV), tree_mask=tm).cpu().float()
    assert torch.allclose(got, want.float(), atol=2e-2), \
        (got - want).abs().max()
