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


def test_server_adapter_preload_swarm(tmp_path):
    """Adapters preloaded at server start (run_server --adapters layout):
    a client selecting the adapter gets different-but-deterministic output;
    without selection the base output is unchanged (exact)."""
    import torch

    from bloombee_amd.client import ClientConfig
    from bloombee_amd.engine import BlockStack, LocalEngine
    from bloombee_amd.models.auto import AutoDistributedModelForCausalLM
    from bloombee_amd.models.base import resolve_config
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server

    cfg_m = resolve_config("llama-tiny")
    # author an adapter per block, saved in the per-block layout
    stack = BlockStack(cfg_m, 0, 4, device="cpu", seed=0)
    torch.manual_seed(41)
    for i, blk in enumerate(stack.blocks):
        sets = create_lora_adapter(blk, rank=2, seed=40 + i)
        for s in sets.values():
            s.b.normal_(0, 0.05)  # nonzero so the delta actually bites
        save_adapter(sets, str(tmp_path / f"block{i}"))

    boot = Dht()
    srv = {"adapters": {"demo": str(tmp_path)}}
    s1 = Server("llama-tiny", initial_peers=[boot.endpoint],
                block_indices=(0, 2), device="cpu", seed=0,
                kv_max_tokens=1 << 14, update_period=5.0, **srv)
    s2 = Server("llama-tiny", initial_peers=[boot.endpoint],
                block_indices=(2, 4), device="cpu", seed=0,
                kv_max_tokens=1 << 14, update_period=5.0, **srv)
    s1.run_in_background()
    s2.run_in_background()
    try:
        gen = torch.Generator().manual_seed(3)
        prompt = torch.randint(0, 1000, (1, 6), generator=gen)

        base_cfg = ClientConfig(initial_peers=[boot.endpoint])
        base = AutoDistributedModelForCausalLM.from_pretrained(
            "llama-tiny", client_config=base_cfg, seed=0)
        out_base = base.generate(prompt, max_new_tokens=5)
        base.remote.manager.shutdown()

        eng = LocalEngine("llama-tiny", device="cpu", seed=0,
                          kv_max_tokens=1 << 14)
        kv = eng.kv_pool.allocate(1, 64)
        toks = [eng.prefill(prompt, kv)]
        for _ in range(4):
            toks.append(eng.decode_step(toks[-1], kv))
        kv.close()
        assert torch.equal(out_base[:, 6:], torch.stack(toks, 1))

        ad_cfg = ClientConfig(initial_peers=[boot.endpoint],
                              active_adapter="demo")
        ad = AutoDistributedModelForCausalLM.from_pretrained(
            "llama-tiny", client_config=ad_cfg, seed=0)
        out_ad = ad.generate(prompt, max_new_tokens=5)
        out_ad2 = ad.generate(prompt, max_new_tokens=5)
        assert torch.equal(out_ad, out_ad2)  # deterministic
        assert not torch.equal(out_ad, out_base)  # adapter changes the dist
        ad.remote.manager.shutdown()
    finally:
        s1.shutdown()
        s2.shutdown()
        boot.shutdown()


def test_training_forward_respects_adapter(tmp_path):
    """rpc_forward/backward run under the client's active adapter (the
    training path, not just inference sessions)."""
    import torch

    from bloombee_amd.client import ClientConfig
    from bloombee_amd.engine import BlockStack
    from bloombee_amd.models.auto import AutoDistributedModelForCausalLM
    from bloombee_amd.models.base import resolve_config
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server

    cfg_m = resolve_config("llama-tiny")
    stack = BlockStack(cfg_m, 0, 4, device="cpu", seed=0)
    torch.manual_seed(51)
    for i, blk in enumerate(stack.blocks):
        sets = create_lora_adapter(blk, rank=2, seed=50 + i)
        for s in sets.values():
            s.b.normal_(0, 0.05)
        save_adapter(sets, str(tmp_path / f"block{i}"))

    boot = Dht()
    s1 = Server("llama-tiny", initial_peers=[boot.endpoint],
                block_indices=(0, 4), device="cpu", seed=0,
                kv_max_tokens=1 << 14, update_period=5.0,
                adapters={"demo": str(tmp_path)})
    s1.run_in_background()
    try:
        ids = torch.randint(0, 1000, (1, 5),
                            generator=torch.Generator().manual_seed(2))
        base = AutoDistributedModelForCausalLM.from_pretrained(
            "llama-tiny",
            client_config=ClientConfig(initial_peers=[boot.endpoint]), seed=0)
        h_base = base.transformer.remote(base.transformer.embed(ids))
        base.remote.manager.shutdown()

        ad = AutoDistributedModelForCausalLM.from_pretrained(
            "llama-tiny",
            client_config=ClientConfig(initial_peers=[boot.endpoint],
                                       active_adapter="demo"), seed=0)
        h = ad.transformer.embed(ids).detach().requires_grad_(True)
        out = ad.transformer.remote(h)
        assert not torch.equal(out, h_base)  # adapter changes training fwd
        out.float().square().mean().backward()
        assert h.grad is not None and torch.isfinite(h.grad).all()
        ad.remote.manager.shutdown()
    finally:
        s1.shutdown()
        boot.shutdown()


def test_adapter_plus_deep_ptune_compose(tmp_path):
    """LoRA adapters (server-side, frozen) and deep prompt-tuning
    (client-side, trainable) must compose: prompts train against the
    adapter-modified stack."""
    import torch

    from bloombee_amd.client import ClientConfig
    from bloombee_amd.engine import BlockStack
    from bloombee_amd.models.auto import AutoDistributedModelForCausalLM
    from bloombee_amd.models.base import resolve_config
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server

    cfg_m = resolve_config("llama-tiny")
    stack = BlockStack(cfg_m, 0, 4, device="cpu", seed=0)
    torch.manual_seed(61)
    for i, blk in enumerate(stack.blocks):
        sets = create_lora_adapter(blk, rank=2, seed=60 + i)
        for s in sets.values():
            s.b.normal_(0, 0.05)
        save_adapter(sets, str(tmp_path / f"block{i}"))

    boot = Dht()
    srv = Server("llama-tiny", initial_peers=[boot.endpoint],
                 block_indices=(0, 4), device="cpu", seed=0,
                 kv_max_tokens=1 << 14, adapters={"demo": str(tmp_path)})
    srv.run_in_background()
    try:
        model = AutoDistributedModelForCausalLM.from_pretrained(
            "llama-tiny",
            client_config=ClientConfig(initial_peers=[boot.endpoint],
                                       active_adapter="demo"),
            seed=0, pre_seq_len=4, deep_ptune=True)
        opt = torch.optim.Adam(model.trainable_parameters(), lr=5e-2)
        gen = torch.Generator().manual_seed(62)
        ids = torch.randint(0, 1000, (1, 6), generator=gen)
        target = torch.randint(0, 1000, (1, 10), generator=gen)
        losses = []
        for _ in range(3):
            logits = model(ids)
            loss = torch.nn.functional.cross_entropy(
                logits.float().view(-1, logits.shape[-1]), target.view(-1))
            opt.zero_grad()
            loss.backward()
            opt.step()
            losses.append(float(loss.detach()))
        assert losses[-1] < losses[0], losses
        model.remote.manager.shutdown()
    finally:
        srv.shutdown()
        boot.shutdown()
