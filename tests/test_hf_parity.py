"""Numerical parity vs HuggingFace transformers blocks with copied weights
(the reference's strongest parity tier: test_optimized_layers.py /
test_qwen3_block_parity.py — random weights, fp32, atol 1e-4 class)."""
import pytest
import torch

from bloombee_amd.models.base import resolve_config


def _mk_block(model, seed=3):
    from bloombee_amd.engine import BlockStack

    cfg = resolve_config(model)
    cfg.torch_dtype = "float32"
    stack = BlockStack(cfg, 0, 1, device="cpu", seed=seed)
    return cfg, stack.blocks[0], stack


def test_llama_block_matches_hf():
    from transformers.models.llama.configuration_llama import LlamaConfig as HFCfg
    from transformers.models.llama.modeling_llama import (LlamaDecoderLayer,
                                                          LlamaRotaryEmbedding)

    cfg, blk, stack = _mk_block("llama-tiny")
    hf_cfg = HFCfg(
        hidden_size=cfg.hidden_size, intermediate_size=cfg.intermediate_size,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        head_dim=cfg.head_dim, rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta, attention_bias=False, attention_dropout=0.0,
        max_position_embeddings=cfg.max_position_embeddings,
        vocab_size=cfg.vocab_size, attn_implementation="eager",
    )
    layer = LlamaDecoderLayer(hf_cfg, layer_idx=0).eval()
    Hq, Hkv, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
    with torch.no_grad():
        layer.input_layernorm.weight.copy_(blk.input_norm_w)
        layer.post_attention_layernorm.weight.copy_(blk.post_norm_w)
        qkv = blk.qkv_w
        layer.self_attn.q_proj.weight.copy_(qkv[:Hq * D])
        layer.self_attn.k_proj.weight.copy_(qkv[Hq * D:(Hq + Hkv) * D])
        layer.self_attn.v_proj.weight.copy_(qkv[(Hq + Hkv) * D:])
        layer.self_attn.o_proj.weight.copy_(blk.o_w)
        I = cfg.intermediate_size
        layer.mlp.gate_proj.weight.copy_(blk.gate_up_w[:I])
        layer.mlp.up_proj.weight.copy_(blk.gate_up_w[I:])
        layer.mlp.down_proj.weight.copy_(blk.down_w)

    torch.manual_seed(0)
    B, T = 2, 11
    h = torch.randn(B, T, cfg.hidden_size) * 0.3
    rotary = LlamaRotaryEmbedding(hf_cfg)
    pos = torch.arange(T).unsqueeze(0).expand(B, T)
    cos_sin = rotary(h, pos)
    mask = torch.full((T, T), float("-inf")).triu(1).view(1, 1, T, T).expand(B, 1, T, T)
    with torch.no_grad():
        hf_out = layer(h, attention_mask=mask, position_ids=pos,
                       position_embeddings=cos_sin)
        if isinstance(hf_out, tuple):
            hf_out = hf_out[0]

    kv = stack.make_kv(1024).allocate(B, 64)
    kv.extend(T)
    ours = blk.forward_inference(h.clone(), kv,
                                 torch.zeros(B, dtype=torch.int32))
    kv.close()
    diff = (ours - hf_out).abs().max().item()
    assert diff < 1e-4, f"llama block vs HF: max diff {diff}"


def test_qwen3_block_matches_hf():
    from transformers.models.qwen3.configuration_qwen3 import Qwen3Config as HFCfg
    from transformers.models.qwen3.modeling_qwen3 import (Qwen3DecoderLayer,
                                                          Qwen3RotaryEmbedding)

    cfg, blk, stack = _mk_block("qwen3-tiny")
    hf_cfg = HFCfg(
        hidden_size=cfg.hidden_size, intermediate_size=cfg.intermediate_size,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        head_dim=cfg.head_dim, rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta, attention_bias=False, attention_dropout=0.0,
        max_position_embeddings=cfg.max_position_embeddings,
        vocab_size=cfg.vocab_size, attn_implementation="eager",
    )
    layer = Qwen3DecoderLayer(hf_cfg, layer_idx=0).eval()
    Hq, Hkv, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
    with torch.no_grad():
        layer.input_layernorm.weight.copy_(blk.input_norm_w)
        layer.post_attention_layernorm.weight.copy_(blk.post_norm_w)
        qkv = blk.qkv_w
        layer.self_attn.q_proj.weight.copy_(qkv[:Hq * D])
        layer.self_attn.k_proj.weight.copy_(qkv[Hq * D:(Hq + Hkv) * D])
        layer.self_attn.v_proj.weight.copy_(qkv[(Hq + Hkv) * D:])
        layer.self_attn.o_proj.weight.copy_(blk.o_w)
        layer.self_attn.q_norm.weight.copy_(blk.q_norm_w)
        layer.self_attn.k_norm.weight.copy_(blk.k_norm_w)
        I = cfg.intermediate_size
        layer.mlp.gate_proj.weight.copy_(blk.gate_up_w[:I])
        layer.mlp.up_proj.weight.copy_(blk.gate_up_w[I:])
        layer.mlp.down_proj.weight.copy_(blk.down_w)

    torch.manual_seed(0)
    B, T = 2, 9
    h = torch.randn(B, T, cfg.hidden_size) * 0.3
    rotary = Qwen3RotaryEmbedding(hf_cfg)
    pos = torch.arange(T).unsqueeze(0).expand(B, T)
    cos_sin = rotary(h, pos)
    mask = torch.full((T, T), float("-inf")).triu(1).view(1, 1, T, T).expand(B, 1, T, T)
    with torch.no_grad():
        hf_out = layer(h, attention_mask=mask, position_ids=pos,
                       position_embeddings=cos_sin)
        if isinstance(hf_out, tuple):
            hf_out = hf_out[0]

    kv = stack.make_kv(1024).allocate(B, 64)
    kv.extend(T)
    ours = blk.forward_inference(h.clone(), kv,
                                 torch.zeros(B, dtype=torch.int32))
    kv.close()
    diff = (ours - hf_out).abs().max().item()
    assert diff < 1e-4, f"qwen3 block vs HF: max diff {diff}"


def test_mixtral_block_matches_hf():
    from transformers.models.mixtral.configuration_mixtral import \
        MixtralConfig as HFCfg
    from transformers.models.mixtral.modeling_mixtral import (
        MixtralDecoderLayer, MixtralRotaryEmbedding)

    cfg, blk, stack = _mk_block("mixtral-tiny")
    hf_cfg = HFCfg(
        hidden_size=cfg.hidden_size, intermediate_size=cfg.intermediate_size,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        rms_norm_eps=cfg.rms_norm_eps, rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        vocab_size=cfg.vocab_size, num_local_experts=blk.E,
        num_experts_per_tok=blk.topk, attn_implementation="eager",
        attention_dropout=0.0,
    )
    layer = MixtralDecoderLayer(hf_cfg, layer_idx=0).eval()
    Hq, Hkv, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
    I = cfg.intermediate_size
    with torch.no_grad():
        layer.input_layernorm.weight.copy_(blk.input_norm_w)
        layer.post_attention_layernorm.weight.copy_(blk.post_norm_w)
        qkv = blk.qkv_w
        layer.self_attn.q_proj.weight.copy_(qkv[:Hq * D])
        layer.self_attn.k_proj.weight.copy_(qkv[Hq * D:(Hq + Hkv) * D])
        layer.self_attn.v_proj.weight.copy_(qkv[(Hq + Hkv) * D:])
        layer.self_attn.o_proj.weight.copy_(blk.o_w)
        mlp = layer.block_sparse_moe if hasattr(layer, "block_sparse_moe") else layer.mlp
        (mlp.gate if hasattr(mlp, "gate") else mlp.router).weight.copy_(blk.router_w)
        experts = mlp.experts
        if hasattr(experts, "gate_up_proj"):  # transformers 5.x batched experts
            experts.gate_up_proj.copy_(blk.expert_gate_up_w)
            experts.down_proj.copy_(blk.expert_down_w)
        else:
            for e in range(blk.E):
                ex = experts[e]
                ex.w1.weight.copy_(blk.expert_gate_up_w[e][:I])
                ex.w3.weight.copy_(blk.expert_gate_up_w[e][I:])
                ex.w2.weight.copy_(blk.expert_down_w[e])

    torch.manual_seed(0)
    B, T = 2, 9
    h = torch.randn(B, T, cfg.hidden_size) * 0.3
    rotary = MixtralRotaryEmbedding(hf_cfg)
    pos = torch.arange(T).unsqueeze(0).expand(B, T)
    cos_sin = rotary(h, pos)
    mask = torch.full((T, T), float("-inf")).triu(1).view(1, 1, T, T).expand(B, 1, T, T)
    with torch.no_grad():
        hf_out = layer(h, attention_mask=mask, position_ids=pos,
                       position_embeddings=cos_sin)
        if isinstance(hf_out, tuple):
            hf_out = hf_out[0]

    kv = stack.make_kv(1024).allocate(B, 64)
    kv.extend(T)
    ours = blk.forward_inference(h.clone(), kv,
                                 torch.zeros(B, dtype=torch.int32))
    kv.close()
    diff = (ours - hf_out).abs().max().item()
    assert diff < 1e-4, f"mixtral block vs HF: max diff {diff}"


def test_gemma4_block_matches_hf():
    from transformers.models.gemma4.configuration_gemma4 import Gemma4TextConfig
    from transformers.models.gemma4.modeling_gemma4 import (
        Gemma4TextDecoderLayer, Gemma4TextRotaryEmbedding)

    cfg, blk, stack = _mk_block("gemma4-tiny")
    li = 0  # sliding layer
    hf_cfg = Gemma4TextConfig(
        hidden_size=cfg.hidden_size, num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        intermediate_size=cfg.intermediate_size, vocab_size=cfg.vocab_size,
        head_dim=cfg.head_dim,
        global_head_dim=int(cfg.extras["global_head_dim"]),
        hidden_size_per_layer_input=0, sliding_window=int(cfg.extras["sliding_window"]),
        rms_norm_eps=cfg.rms_norm_eps, attn_implementation="eager",
        attention_dropout=0.0,
    )
    layer = Gemma4TextDecoderLayer(hf_cfg, layer_idx=li).eval()
    Hq, Hkv, D = blk.Hq, blk.Hkv, blk.D
    with torch.no_grad():
        layer.input_layernorm.weight.copy_(blk.input_norm_w)
        layer.post_attention_layernorm.weight.copy_(blk.post_attn_norm_w)
        layer.pre_feedforward_layernorm.weight.copy_(blk.pre_ffn_norm_w)
        layer.post_feedforward_layernorm.weight.copy_(blk.post_ffn_norm_w)
        layer.self_attn.q_proj.weight.copy_(blk.q_w)
        layer.self_attn.k_proj.weight.copy_(blk.k_w)
        layer.self_attn.v_proj.weight.copy_(blk.v_w)
        layer.self_attn.o_proj.weight.copy_(blk.o_w)
        layer.self_attn.q_norm.weight.copy_(blk.q_norm_w)
        layer.self_attn.k_norm.weight.copy_(blk.k_norm_w)
        I = cfg.intermediate_size
        layer.mlp.gate_proj.weight.copy_(blk.gate_up_w[:I])
        layer.mlp.up_proj.weight.copy_(blk.gate_up_w[I:])
        layer.mlp.down_proj.weight.copy_(blk.down_w)

    torch.manual_seed(0)
    B, T = 1, 12  # > sliding_window=8 so the window path is exercised
    h = torch.randn(B, T, cfg.hidden_size) * 0.3
    rotary = Gemma4TextRotaryEmbedding(hf_cfg)
    pos = torch.arange(T).unsqueeze(0).expand(B, T)
    pos_emb = rotary(h, pos, layer_type="sliding_attention")
    if isinstance(pos_emb, dict):
        pos_emb = pos_emb["sliding_attention"]
    qpos = torch.arange(T).view(T, 1)
    kpos = torch.arange(T).view(1, T)
    w = int(cfg.extras["sliding_window"])
    keep = (kpos <= qpos) & (kpos > qpos - w)
    mask = torch.where(keep, 0.0, float("-inf")).view(1, 1, T, T)
    with torch.no_grad():
        hf_out = layer(h, shared_kv_states={}, position_embeddings=pos_emb,
                       attention_mask=mask, position_ids=pos)
        if isinstance(hf_out, tuple):
            hf_out = hf_out[0]

    kv = stack.make_kv(1024).allocate(B, 64)
    kv.extend(T)
    ours = blk.forward_inference(h.clone(), kv,
                                 torch.zeros(B, dtype=torch.int32))
    kv.close()
    diff = (ours - hf_out).abs().max().item()
    assert diff < 1e-4, f"gemma4 block vs HF: max diff {diff}"


def test_bloom_block_matches_hf():
    from transformers.models.bloom.configuration_bloom import BloomConfig as HFCfg
    from transformers.models.bloom.modeling_bloom import (BloomBlock,
                                                          build_alibi_tensor)

    cfg, blk, stack = _mk_block("bloom-tiny")
    Hq, D, H = blk.Hq, blk.D, cfg.hidden_size
    hf_cfg = HFCfg(hidden_size=H, n_head=Hq,
                   n_layer=cfg.num_hidden_layers, vocab_size=cfg.vocab_size,
                   layer_norm_epsilon=cfg.layer_norm_epsilon,
                   attn_implementation="eager", hidden_dropout=0.0,
                   attention_dropout=0.0)
    layer = BloomBlock(hf_cfg, layer_idx=0).eval()
    with torch.no_grad():
        layer.input_layernorm.weight.copy_(blk.ln1_w)
        layer.input_layernorm.bias.copy_(blk.ln1_b)
        layer.post_attention_layernorm.weight.copy_(blk.ln2_w)
        layer.post_attention_layernorm.bias.copy_(blk.ln2_b)
        # HF bloom fuses qkv per-head interleaved [h: q|k|v]; ours is q|k|v
        qkv_hf = torch.empty_like(blk.qkv_w)
        qkvb_hf = torch.empty_like(blk.qkv_b)
        for h in range(Hq):
            for j, sec in enumerate(range(3)):
                qkv_hf[(h * 3 + j) * D:(h * 3 + j + 1) * D] = \
                    blk.qkv_w[j * Hq * D + h * D: j * Hq * D + (h + 1) * D]
                qkvb_hf[(h * 3 + j) * D:(h * 3 + j + 1) * D] = \
                    blk.qkv_b[j * Hq * D + h * D: j * Hq * D + (h + 1) * D]
        layer.self_attention.query_key_value.weight.copy_(qkv_hf)
        layer.self_attention.query_key_value.bias.copy_(qkvb_hf)
        layer.self_attention.dense.weight.copy_(blk.dense_w)
        layer.self_attention.dense.bias.copy_(blk.dense_b)
        layer.mlp.dense_h_to_4h.weight.copy_(blk.up_w)
        layer.mlp.dense_h_to_4h.bias.copy_(blk.up_b)
        layer.mlp.dense_4h_to_h.weight.copy_(blk.down_w)
        layer.mlp.dense_4h_to_h.bias.copy_(blk.down_b)

    torch.manual_seed(0)
    B, T = 2, 9
    h = torch.randn(B, T, H) * 0.3
    mask2d = torch.ones(B, T, dtype=torch.long)
    alibi = build_alibi_tensor(mask2d, Hq, dtype=torch.float32)
    causal = torch.full((T, T), float("-inf")).triu(1).view(1, 1, T, T).expand(B, 1, T, T)
    with torch.no_grad():
        hf_out = layer(h, alibi=alibi, attention_mask=causal)
        if isinstance(hf_out, tuple):
            hf_out = hf_out[0]

    kv = stack.make_kv(1024).allocate(B, 64)
    kv.extend(T)
    ours = blk.forward_inference(h.clone(), kv,
                                 torch.zeros(B, dtype=torch.int32))
    kv.close()
    diff = (ours - hf_out).abs().max().item()
    assert diff < 1e-4, f"bloom block vs HF: max diff {diff}"


def test_falcon_block_matches_hf():
    from transformers.models.falcon.configuration_falcon import \
        FalconConfig as HFCfg
    from transformers.models.falcon.modeling_falcon import (
        FalconDecoderLayer, FalconRotaryEmbedding)

    cfg, blk, stack = _mk_block("falcon-tiny")
    Hq, Hkv, D, H = blk.Hq, blk.Hkv, blk.D, cfg.hidden_size
    hf_cfg = HFCfg(hidden_size=H, num_attention_heads=Hq,
                   num_hidden_layers=cfg.num_hidden_layers,
                   vocab_size=cfg.vocab_size, multi_query=True,
                   parallel_attn=True, bias=False,
                   new_decoder_architecture=False,
                   layer_norm_epsilon=cfg.layer_norm_epsilon,
                   rope_theta=cfg.rope_theta,
                   attn_implementation="eager", hidden_dropout=0.0,
                   attention_dropout=0.0)
    layer = FalconDecoderLayer(hf_cfg, layer_idx=0).eval()
    with torch.no_grad():
        layer.input_layernorm.weight.copy_(blk.ln_w)
        layer.input_layernorm.bias.copy_(blk.ln_b)
        layer.self_attention.query_key_value.weight.copy_(blk.qkv_w)
        layer.self_attention.dense.weight.copy_(blk.o_w)
        layer.mlp.dense_h_to_4h.weight.copy_(blk.up_w)
        layer.mlp.dense_4h_to_h.weight.copy_(blk.down_w)

    torch.manual_seed(0)
    B, T = 2, 9
    h = torch.randn(B, T, H) * 0.3
    rotary = FalconRotaryEmbedding(hf_cfg)
    pos = torch.arange(T).unsqueeze(0).expand(B, T)
    cos_sin = rotary(h, pos)
    mask = torch.full((T, T), float("-inf")).triu(1).view(1, 1, T, T).expand(B, 1, T, T)
    with torch.no_grad():
        hf_out = layer(h, position_embeddings=cos_sin, attention_mask=mask,
                       alibi=None, position_ids=pos)
        if isinstance(hf_out, tuple):
            hf_out = hf_out[0]

    kv = stack.make_kv(1024).allocate(B, 64)
    kv.extend(T)
    ours = blk.forward_inference(h.clone(), kv,
                                 torch.zeros(B, dtype=torch.int32))
    kv.close()
    diff = (ours - hf_out).abs().max().item()
    assert diff < 1e-4, f"falcon block vs HF: max diff {diff}"
