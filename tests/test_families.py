"""Per-family block tests: decode==prefill consistency, inference==train-path
numerics, and end-to-end greedy sanity on CPU (mirrors the reference's
block-parity tier, SURVEY.md §4 cat 2)."""
import pytest
import torch

from bloombee_amd.engine import LocalEngine

FAMILIES = ["llama-tiny", "bloom-tiny", "falcon-tiny", "qwen3-tiny",
            "mixtral-tiny", "gemma4-tiny"]


@pytest.mark.parametrize("model", FAMILIES)
def test_decode_equals_prefill(model):
    torch.manual_seed(0)
    eng = LocalEngine(model, device="cpu", seed=1, kv_max_tokens=4096)
    ids = torch.randint(0, min(900, eng.config.vocab_size), (2, 9),
                        generator=torch.Generator().manual_seed(3))
    # full prefill
    kv1 = eng.kv_pool.allocate(2, 64)
    t_full = eng.prefill(ids, kv1)
    kv1.close()
    # token-by-token decode after 1-token prefill
    kv2 = eng.kv_pool.allocate(2, 64)
    tok = eng.prefill(ids[:, :1], kv2)
    for t in range(1, 9):
        tok = eng.decode_step(ids[:, t], kv2)
    kv2.close()
    assert torch.equal(t_full, tok), f"{model}: decode path != prefill path"


@pytest.mark.parametrize("model", FAMILIES)
def test_train_path_matches_inference(model):
    torch.manual_seed(0)
    eng = LocalEngine(model, device="cpu", seed=1, kv_max_tokens=4096)
    B, T = 2, 8
    h = (torch.randn(B, T, eng.config.hidden_size,
                     generator=torch.Generator().manual_seed(4)) * 0.1
         ).to(eng.config.dtype)
    kv = eng.kv_pool.allocate(B, 32)
    kv.extend(T)
    start = torch.zeros(B, dtype=torch.int32)
    out_inf = eng.stack.forward_inference(h.clone(), kv, start)
    kv.close()
    out_train = eng.stack.forward_train(h.clone())
    assert torch.allclose(out_inf.float(), out_train.float(), atol=5e-2), \
        f"{model}: max diff {(out_inf.float()-out_train.float()).abs().max()}"


@pytest.mark.parametrize("model", FAMILIES)
def test_generate_runs(model):
    eng = LocalEngine(model, device="cpu", seed=0, kv_max_tokens=4096)
    ids = torch.randint(0, min(900, eng.config.vocab_size), (1, 5),
                        generator=torch.Generator().manual_seed(1))
    out = eng.generate_greedy(ids, 4)
    assert out.shape == (1, 4)
    assert (out >= 0).all() and (out < eng.config.vocab_size).all()
