"""LoRA adapter tests (mirror reference test_peft.py)."""
import torch

from bloombee_amd.engine import BlockStack
from bloombee_amd.models.base import resolve_config
from bloombee_amd.utils.peft import (add_adapter_to_block, create_lora_adapter,
                                     estimate_adapter_memory, load_adapter,
                                     save_adapter, using_adapter)


def _fwd(stack, x):
    kv = stack.make_kv(1024)
    h = kv.allocate(2, 32)
    h.extend(5)
    out = stack.blocks[0].forward_inference(x, h, torch.zeros(2, dtype=torch.int32))
    h.close()
    return out


def test_lora_zero_b_is_identity_and_trains():
    cfg = resolve_config("llama-tiny")
    stack = BlockStack(cfg, 0, 1, device="cpu", seed=2)
    blk = stack.blocks[0]
    x = (torch.randn(2, 5, cfg.hidden_size,
                     generator=torch.Generator().manual_seed(1)) * 0.1).to(cfg.dtype)
    base = _fwd(stack, x)
    sets = create_lora_adapter(blk, rank=4, seed=0)
    add_adapter_to_block(blk, "demo", sets)
    with using_adapter("demo"):
        out = _fwd(stack, x)
    assert torch.equal(out, base)  # B starts at zero -> exact identity
    # non-zero B changes the output only inside the context
    with torch.no_grad():
        sets["qkv_w"].b.add_(torch.randn_like(sets["qkv_w"].b) * 0.1)
    with using_adapter("demo"):
        changed = _fwd(stack, x)
    assert not torch.equal(changed, base)
    assert torch.equal(_fwd(stack, x), base)  # no adapter active -> base


def test_adapter_save_load_roundtrip(tmp_path):
    cfg = resolve_config("llama-tiny")
    stack = BlockStack(cfg, 0, 1, device="cpu", seed=2)
    sets = create_lora_adapter(stack.blocks[0], rank=4, seed=1)
    with torch.no_grad():
        sets["o_w"].b.add_(0.05)
    save_adapter(sets, str(tmp_path / "ad"))
    back = load_adapter(str(tmp_path / "ad"), dtype=torch.bfloat16)
    assert set(back) == set(sets)
    assert torch.allclose(back["o_w"].b.float(), sets["o_w"].b.float(), atol=1e-2)
    assert estimate_adapter_memory(stack.blocks[0], 4) > 0
