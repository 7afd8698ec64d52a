"""CPU reference op sanity: shapes, math identities, and known-good cross-checks.
These are the goldens the GPU parity suite (test_gpu_kernels.py) compares to."""
import math

import pytest
import torch

from bloombee_amd.ops import reference as ref


def test_rms_norm_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(3, 5, 64)
    w = torch.randn(64)
    y = ref.rms_norm(x, w, eps=1e-5)
    expect = x / (x.pow(2).mean(-1, keepdim=True) + 1e-5).sqrt() * w
    assert torch.allclose(y, expect, atol=1e-5)


def test_rms_norm_residual_fuses_add():
    x = torch.randn(2, 4, 32)
    r = torch.randn(2, 4, 32)
    w = torch.ones(32)
    h, y = ref.rms_norm_residual(x, r, w)
    assert torch.allclose(h, x + r, atol=1e-6)
    assert torch.allclose(y, ref.rms_norm(x + r, w), atol=1e-6)


def test_rope_inverse_rotation():
    """Rotating by pos p then by -p (via conjugate) returns the original."""
    D = 32
    cos, sin = ref.rope_cos_sin(D, 128)
    q = torch.randn(1, 2, 4, D)
    k = torch.randn(1, 1, 4, D)
    pos = torch.arange(4).view(1, 4)
    q1, k1 = ref.rope_apply(q, k, cos, sin, pos)
    q2, _ = ref.rope_apply(q1, k1, cos, -sin, pos)
    assert torch.allclose(q2, q, atol=1e-5)


def test_rope_dot_product_depends_on_relative_position():
    D = 64
    cos, sin = ref.rope_cos_sin(D, 256)
    v = torch.randn(1, 1, 1, D)
    def rot(p):
        q, _ = ref.rope_apply(v, v, cos, sin, torch.tensor([[p]]))
        return q.flatten()
    d1 = torch.dot(rot(3), rot(7))
    d2 = torch.dot(rot(13), rot(17))
    assert abs(d1 - d2) < 1e-3 * v.norm() ** 2


def test_swiglu():
    g = torch.randn(4, 16)
    u = torch.randn(4, 16)
    out = ref.swiglu(g, u)
    assert torch.allclose(out, torch.nn.functional.silu(g) * u, atol=1e-5)


def test_attn_paged_matches_sdpa_prefill():
    torch.manual_seed(1)
    B, Hq, Hkv, T, D, P = 2, 4, 2, 23, 32, 16
    npages = 64
    kp = torch.zeros(npages, Hkv, P, D)
    vp = torch.zeros(npages, Hkv, D, P)
    maxp = 4
    pt = torch.arange(B * maxp, dtype=torch.int32).reshape(B, maxp)
    q = torch.randn(B, Hq, T, D)
    k = torch.randn(B, Hkv, T, D)
    v = torch.randn(B, Hkv, T, D)
    start = torch.zeros(B, dtype=torch.int32)
    ref.kv_write(k, v, kp, vp, pt, start)
    out = ref.attn_paged(q, kp, vp, pt, start.long())
    # dense reference
    G = Hq // Hkv
    kd = k.repeat_interleave(G, 1)
    vd = v.repeat_interleave(G, 1)
    mask = torch.ones(T, T, dtype=torch.bool).tril()
    expect = torch.nn.functional.scaled_dot_product_attention(
        q.float(), kd.float(), vd.float(), attn_mask=mask)
    assert torch.allclose(out, expect, atol=1e-4), (out - expect).abs().max()


def test_attn_paged_decode_step_matches_full():
    """Decode (Tq=1) after a prefill must equal the last row of full attention."""
    torch.manual_seed(2)
    B, Hq, Hkv, D, P = 1, 4, 4, 16, 16
    kp = torch.zeros(8, Hkv, P, D)
    vp = torch.zeros(8, Hkv, D, P)
    pt = torch.arange(8, dtype=torch.int32).reshape(1, 8)
    T = 21
    q = torch.randn(B, Hq, T, D)
    k = torch.randn(B, Hkv, T, D)
    v = torch.randn(B, Hkv, T, D)
    z = torch.zeros(B, dtype=torch.int32)
    ref.kv_write(k, v, kp, vp, pt, z)
    full = ref.attn_paged(q, kp, vp, pt, z.long())
    dec = ref.attn_paged(q[:, :, -1:], kp, vp, pt,
                         torch.tensor([T - 1], dtype=torch.long))
    assert torch.allclose(dec, full[:, :, -1:], atol=1e-5)


def test_attn_sliding_window():
    torch.manual_seed(3)
    B, H, T, D, P = 1, 2, 40, 16, 16
    kp = torch.zeros(8, H, P, D)
    vp = torch.zeros(8, H, D, P)
    pt = torch.arange(8, dtype=torch.int32).reshape(1, 8)
    q = torch.randn(B, H, T, D)
    k = torch.randn(B, H, T, D)
    v = torch.randn(B, H, T, D)
    z = torch.zeros(B, dtype=torch.int32)
    ref.kv_write(k, v, kp, vp, pt, z)
    win = 8
    out = ref.attn_paged(q, kp, vp, pt, z.long(), sliding_window=win)
    # manual mask
    scores = torch.einsum("htd,hcd->htc", q[0].float(), k[0].float()) / math.sqrt(D)
    pos_q = torch.arange(T).view(T, 1)
    pos_k = torch.arange(T).view(1, T)
    mask = (pos_k <= pos_q) & (pos_k > pos_q - win)
    scores = scores.masked_fill(~mask, float("-inf"))
    expect = torch.einsum("htc,hcd->htd", scores.softmax(-1), v[0].float())
    assert torch.allclose(out[0], expect, atol=1e-5)


def test_tree_mask_attention():
    """Tree attention: each draft token attends to its ancestors only
    (spec-decode verify path, ref backend.py:944-1047)."""
    torch.manual_seed(4)
    B, H, D, P = 1, 2, 16, 16
    kp = torch.zeros(8, H, P, D)
    vp = torch.zeros(8, H, D, P)
    pt = torch.arange(8, dtype=torch.int32).reshape(1, 8)
    Tpre, Ttree = 5, 3
    k = torch.randn(B, H, Tpre + Ttree, D)
    v = torch.randn(B, H, Tpre + Ttree, D)
    q = torch.randn(B, H, Ttree, D)
    z = torch.zeros(B, dtype=torch.int32)
    ref.kv_write(k, v, kp, vp, pt, z)
    # tree: token0 root; token1, token2 both children of token0 (siblings)
    tm = torch.tensor([[[1, 0, 0], [1, 1, 0], [1, 0, 1]]], dtype=torch.bool)
    out = ref.attn_paged(q, kp, vp, pt, torch.tensor([Tpre], dtype=torch.long),
                         tree_mask=tm)
    # sibling 2 must NOT see token 1: compare vs manual
    kf = k[0].float()
    vf = v[0].float()
    qf = q[0].float()
    for t, vis in [(1, [0, 1, 2, 3, 4, 5, 6]), (2, [0, 1, 2, 3, 4, 5, 7])]:
        sc = torch.einsum("hd,hcd->hc", qf[:, t], kf[:, vis]) / math.sqrt(D)
        ex = torch.einsum("hc,hcd->hd", sc.softmax(-1), vf[:, vis])
        assert torch.allclose(out[0, :, t], ex, atol=1e-5)


def test_quant4_roundtrip():
    torch.manual_seed(5)
    x = torch.randn(8, 128).to(torch.bfloat16)
    packed, scale, zero = ref.quant4_pack(x)
    y = ref.quant4_unpack(packed, scale, zero)
    err = (x.float() - y.float()).abs().max()
    rng = (x.float().amax(-1) - x.float().amin(-1)).max()
    assert err <= rng / 15 + 0.02


def test_moe_gemm_grouped_reference():
    """Grouped expert GEMM reference vs the straightforward per-token loop
    (rowmap gather, expert grouping, per-slot scale)."""
    torch.manual_seed(9)
    T, K, E, H, N = 6, 2, 4, 32, 64
    A = torch.randn(T, H)
    W = torch.randn(E, N, H)
    experts = torch.randint(0, E, (T, K))
    order = experts.reshape(-1).argsort(stable=True)
    counts = torch.bincount(experts.reshape(-1), minlength=E)
    off = torch.zeros(E + 1, dtype=torch.int32)
    off[1:] = counts.cumsum(0).int()
    tok = (order // K).int()
    scale = torch.rand(T * K)
    C = ref.moe_gemm_grouped(A, W, off, rowmap=tok, scale=scale, S=T * K)
    for s in range(T * K):
        e = int(experts.reshape(-1)[order[s]])
        t = int(order[s]) // K
        want = (A[t].float() @ W[e].float().t()) * scale[s]
        assert torch.allclose(C[s], want, atol=1e-4), s
    # identity rowmap (the down-projection case)
    C2 = ref.moe_gemm_grouped(C, torch.randn(E, 32, N), off, S=T * K)
    assert C2.shape == (T * K, 32)


def test_mixtral_moe_grouped_math_matches_loop():
    """The grouped-MoE host math (sort/bincount/prefix/index_add) composed
    with the reference grouped GEMM equals the per-expert loop output."""
    torch.manual_seed(4)
    T, H, I, E, K = 5, 32, 48, 4, 2
    flat = torch.randn(T, H)
    router_w = torch.randn(E, H) * 0.1
    gu_w = torch.randn(E, 2 * I, H) * 0.1
    dn_w = torch.randn(E, H, I) * 0.1
    logits = (flat @ router_w.t()).float()
    weights, experts = logits.topk(K, -1)
    weights = torch.softmax(weights, -1)
    # loop path
    out_loop = torch.zeros_like(flat)
    for e in range(E):
        sel = (experts == e)
        rows = sel.any(-1).nonzero(as_tuple=True)[0]
        if rows.numel() == 0:
            continue
        xe = flat[rows]
        g, u = (xe @ gu_w[e].t()).split([I, I], -1)
        h = (torch.nn.functional.silu(g.float()).to(u.dtype) * u) @ dn_w[e].t()
        w = (weights.to(flat.dtype) * sel.to(flat.dtype)).sum(-1)[rows]
        out_loop.index_add_(0, rows, h * w.unsqueeze(-1))
    # grouped path
    fe = experts.reshape(-1)
    order = fe.argsort(stable=True)
    counts = torch.bincount(fe, minlength=E)
    off = torch.zeros(E + 1, dtype=torch.int32)
    off[1:] = counts.cumsum(0).int()
    tok = (order // K).int()
    wsorted = weights.reshape(-1)[order].float()
    gu = ref.moe_gemm_grouped(flat, gu_w, off, rowmap=tok, S=T * K)
    g, u = gu.split([I, I], -1)
    act = torch.nn.functional.silu(g.float()).to(u.dtype) * u
    dn = ref.moe_gemm_grouped(act, dn_w, off, scale=wsorted, S=T * K)
    out_grp = torch.zeros_like(flat)
    out_grp.index_add_(0, tok.long(), dn)
    assert torch.allclose(out_grp, out_loop, atol=1e-4), \
        (out_grp - out_loop).abs().max()


def test_linear_w4_fallback_matches_dequant_matmul():
    torch.manual_seed(3)
    N, K, M = 64, 128, 4
    w = torch.randn(N, K)
    x = torch.randn(M, K)
    packed, scale, zero = ref.quant4_pack(w)
    from bloombee_amd import ops
    y = ops.linear_w4(x, packed, scale, zero, N)
    wq = ref.quant4_unpack(packed, scale, zero, dtype=torch.float32).reshape(N, K)
    want = x @ wq.t()
    assert torch.allclose(y, want, atol=1e-4)
    # residual + bias epilogue
    r = torch.randn(M, N)
    b = torch.randn(N)
    y2 = ops.linear_w4(x, packed, scale, zero, N, residual=r, bias=b)
    assert torch.allclose(y2, want + r + b, atol=1e-4)


def test_llama_block_quantized_decode_close_to_full():
    """quantize_weights_q4 keeps the block's decode output within the 4-bit
    quantization tolerance of the full-precision block (CPU fallback path;
    the GPU kernel has its own parity test)."""
    from bloombee_amd.engine import BlockStack
    from bloombee_amd.models.base import resolve_config

    cfg = resolve_config("llama-tiny")
    full = BlockStack(cfg, 0, 1, device="cpu", seed=5)
    quant = BlockStack(cfg, 0, 1, device="cpu", seed=5)
    quant.blocks[0].quantize_weights_q4()
    assert quant.blocks[0].qkv_w.numel() == 0  # bf16 weights dropped
    kvf, kvq = full.make_kv(1 << 10), quant.make_kv(1 << 10)
    hf_, hq_ = kvf.allocate(2, 32), kvq.allocate(2, 32)
    gen = torch.Generator().manual_seed(1)
    x = (torch.randn(2, 4, cfg.hidden_size, generator=gen) * 0.1).to(cfg.dtype)
    sp = torch.zeros(2, dtype=torch.int32)
    hf_.extend(4)
    hq_.extend(4)
    yf = full.forward_inference(x.clone(), hf_, sp)
    yq = quant.forward_inference(x.clone(), hq_, sp)
    # same direction, bounded quant error
    num = (yf.float() - yq.float()).norm()
    den = yf.float().norm().clamp_min(1e-6)
    assert (num / den) < 0.2, (num / den)


def test_qwen3_block_quantized_decode_close_to_full():
    """Qwen3 inherits quantize_weights_q4 from LlamaBlock; its inference
    path must route projections through _lin so the 4-bit weights are the
    ones that run (regression: ops.linear calls bypassed the W4 table)."""
    from bloombee_amd.engine import BlockStack
    from bloombee_amd.models.base import resolve_config

    cfg = resolve_config("qwen3-tiny")
    full = BlockStack(cfg, 0, 1, device="cpu", seed=7)
    quant = BlockStack(cfg, 0, 1, device="cpu", seed=7)
    quant.blocks[0].quantize_weights_q4()
    assert quant.blocks[0].qkv_w.numel() == 0
    kvf, kvq = full.make_kv(1 << 10), quant.make_kv(1 << 10)
    hf_, hq_ = kvf.allocate(2, 32), kvq.allocate(2, 32)
    gen = torch.Generator().manual_seed(2)
    x = (torch.randn(2, 4, cfg.hidden_size, generator=gen) * 0.1).to(cfg.dtype)
    sp = torch.zeros(2, dtype=torch.int32)
    hf_.extend(4)
    hq_.extend(4)
    yf = full.forward_inference(x.clone(), hf_, sp)
    yq = quant.forward_inference(x.clone(), hq_, sp)
    num = (yf.float() - yq.float()).norm()
    den = yf.float().norm().clamp_min(1e-6)
    assert (num / den) < 0.2, (num / den)


def test_linear_norm_fallback_matches_separate():
    """linear(norm=(w,eps)) on CPU == F.linear(rms_norm(x)) exactly (the
    fused GPU kernel has its own parity test in test_gpu_kernels)."""
    from bloombee_amd import ops
    torch.manual_seed(3)
    x = torch.randn(4, 256, dtype=torch.bfloat16)
    w = torch.randn(64, 256, dtype=torch.bfloat16)
    nw = torch.randn(256, dtype=torch.bfloat16)
    want = torch.nn.functional.linear(ops.rms_norm(x, nw, 1e-5), w)
    got = ops.linear(x, w, norm=(nw, 1e-5))
    assert torch.equal(got, want)
    r = torch.randn(4, 64, dtype=torch.bfloat16)
    got2 = ops.linear(x, w, residual=r, norm=(nw, 1e-5))
    assert torch.equal(got2, want + r)


def test_norm_weight_fold_is_reparameterization():
    """fold_norm_weights is math-preserving on every path: the folded block
    (projections *= norm weights, norms = 1) produces the same training
    forward as the original within bf16 fold-rounding, and unfold restores
    the original weights to ~1 ulp."""
    from bloombee_amd.engine import BlockStack
    from bloombee_amd.models.base import resolve_config

    cfg = resolve_config("llama-tiny")
    stack = BlockStack(cfg, 0, 1, device="cpu", seed=11)
    blk = stack.blocks[0]
    # non-trivial norm weights (init_random sets them to 1)
    gen = torch.Generator().manual_seed(3)
    blk.input_norm_w.data = 1.0 + 0.3 * torch.randn(
        cfg.hidden_size, generator=gen).to(cfg.dtype)
    blk.post_norm_w.data = 1.0 + 0.3 * torch.randn(
        cfg.hidden_size, generator=gen).to(cfg.dtype)
    x = (torch.randn(2, 4, cfg.hidden_size, generator=gen) * 0.1).to(cfg.dtype)

    want = blk.forward_train(x.clone())
    orig_qkv = blk.qkv_w.detach().clone()
    orig_inw = blk.input_norm_w.detach().clone()

    blk.fold_norm_weights()
    assert blk._norm_folded
    assert torch.all(blk.input_norm_w == 1.0)
    got = blk.forward_train(x.clone())
    rel = ((got.float() - want.float()).norm()
           / want.float().norm().clamp_min(1e-6)).item()
    assert rel < 2e-2, rel

    blk.unfold_norm_weights()
    assert not blk._norm_folded
    assert torch.equal(blk.input_norm_w, orig_inw)
    drift = ((blk.qkv_w.float() - orig_qkv.float()).abs()
             / orig_qkv.float().abs().clamp_min(1e-3)).max().item()
    assert drift < 2e-2, drift
