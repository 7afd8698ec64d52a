"""Local engine (CPU): decode-vs-prefill consistency and train/infer parity."""
import torch

from bloombee_amd.engine import LocalEngine
from bloombee_amd.models.base import resolve_config


def test_generate_deterministic():
    eng = LocalEngine("llama-tiny", device="cpu", seed=0, kv_max_tokens=4096)
    ids = torch.randint(0, 1000, (2, 7), generator=torch.Generator().manual_seed(0))
    out1 = eng.generate_greedy(ids, 6)
    out2 = eng.generate_greedy(ids, 6)
    assert out1.shape == (2, 6)
    assert torch.equal(out1, out2)


def test_incremental_equals_full_prefill():
    """Feeding the prompt in two chunks must give the same next token as one
    prefill (reference test_full_model.py incremental-session check)."""
    eng = LocalEngine("llama-tiny", device="cpu", seed=1, kv_max_tokens=4096)
    g = torch.Generator().manual_seed(1)
    ids = torch.randint(0, 1000, (1, 12), generator=g)

    kv1 = eng.kv_pool.allocate(1, 64)
    t_full = eng.prefill(ids, kv1)
    kv1.close()

    kv2 = eng.kv_pool.allocate(1, 64)
    eng.prefill(ids[:, :5], kv2)
    t_inc = eng.prefill(ids[:, 5:], kv2)
    kv2.close()
    assert torch.equal(t_full, t_inc)


def test_decode_equals_prefill_token_by_token():
    eng = LocalEngine("llama-tiny", device="cpu", seed=2, kv_max_tokens=4096)
    g = torch.Generator().manual_seed(2)
    ids = torch.randint(0, 1000, (1, 5), generator=g)
    out = eng.generate_greedy(ids, 4)
    # recompute: full prefill over prompt+generated prefix each time
    for i in range(1, 4):
        kv = eng.kv_pool.allocate(1, 64)
        full = torch.cat([ids, out[:, :i]], dim=1)
        nxt = eng.prefill(full, kv)
        kv.close()
        assert torch.equal(nxt, out[:, i]), f"mismatch at step {i}"


def test_train_forward_matches_inference_forward():
    """forward_train (differentiable eager) must match forward_inference
    (paged kernels) on the same block — the two servers paths are numerically
    one model (ref run_rpc_forward vs iterate_rpc_inference)."""
    cfg = resolve_config("llama-tiny")
    cfg.torch_dtype = "float32"
    from bloombee_amd.models.llama.block import LlamaBlock
    from bloombee_amd.kv import PagedKVCache

    blk = LlamaBlock(cfg, 0).init_random(7)
    pool = PagedKVCache(1, cfg.num_key_value_heads, cfg.head_dim,
                        max_tokens=1024, device="cpu", dtype=torch.float32)
    kv = pool.allocate(2, 64)
    T = 9
    x = torch.randn(2, T, cfg.hidden_size)
    start = torch.zeros(2, dtype=torch.int32)
    kv.extend(T)
    y_inf = blk.forward_inference(x, kv, start)
    y_tr = blk.forward_train(x)
    assert torch.allclose(y_inf, y_tr, atol=2e-4), (y_inf - y_tr).abs().max()
    kv.close()


def test_train_forward_grad_flows_to_input():
    cfg = resolve_config("llama-tiny")
    cfg.torch_dtype = "float32"
    from bloombee_amd.models.llama.block import LlamaBlock

    blk = LlamaBlock(cfg, 0).init_random(8)
    x = torch.randn(1, 4, cfg.hidden_size, requires_grad=True)
    y = blk.forward_train(x)
    y.sum().backward()
    assert x.grad is not None and x.grad.abs().sum() > 0
    assert all(p.grad is None for p in blk.parameters())
