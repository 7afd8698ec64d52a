"""Pure-PyTorch fp32 reference implementations of every hot op.

These serve two purposes:
  1. CPU execution path (this container has no GPU; the loopback swarm tests
     and the bloom-560m plumbing config run on CPU).
  2. Numerics golden references for the HIP kernels (tests compare the gfx950
     kernels against these in fp32, mirroring the reference's parity suite,
     e.g. test_mha_gen_llama_decode_parity.py).

Functional parity targets in the reference (ai-decentralized/BloomBee):
  rms_norm           <- flexgen_utils/pytorch_backend.py:111-120
  rope               <- flexgen_utils/pytorch_backend.py:59-110
  attention          <- pytorch_backend.py:665-916 (mha_llama / mha_gen_llama)
  swiglu (mlp core)  <- pytorch_backend.py:1033-1048 (mlp_llama)
  paged KV           <- server/paged_kv.py
  4-bit group quant  <- flexgen_utils/compression.py:94-210

The layouts here are the MI355X-native ones (NOT the reference's):
  hidden      (B, T, H)            bf16
  q           (B, Hq, Tq, D)
  k/v (new)   (B, Hkv, T, D)
  KV pages    (n_pages, Hkv, page_size, D)   -- 4 KB contiguous per
              (page, head) slab at D=128/bf16: one coalesced burst per wave.
  page_table  (B, max_pages_per_seq) int32
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch


# ---------------------------------------------------------------------------
# Normalization
# ---------------------------------------------------------------------------

def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    """RMSNorm with fp32 variance accumulation (ref pytorch_backend.py:111-120)."""
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps)
    return (y * weight.float()).to(x.dtype)


def rms_norm_residual(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Fused residual-add + RMSNorm: h = x + residual; y = rmsnorm(h).

    The fusion target: on MI355X the elementwise add would otherwise be a
    separate HBM round trip (fuse elementwise work into the
    producing kernel).
    """
    h = (x.float() + residual.float()).to(x.dtype)
    return h, rms_norm(h, weight, eps)


def layer_norm(
    x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor], eps: float = 1e-5
) -> torch.Tensor:
    """LayerNorm for bloom/falcon-family blocks (ref pytorch_backend.py mha/mlp)."""
    xf = x.float()
    mu = xf.mean(dim=-1, keepdim=True)
    var = (xf - mu).pow(2).mean(dim=-1, keepdim=True)
    y = (xf - mu) * torch.rsqrt(var + eps)
    y = y * weight.float()
    if bias is not None:
        y = y + bias.float()
    return y.to(x.dtype)


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------

def rope_cos_sin(
    head_dim: int, max_pos: int, theta: float = 10000.0, device=None, dtype=torch.float32,
    scaling: Optional[dict] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Host-precomputed RoPE tables (guide Appendix B: precompute trig on host).

    Returns cos, sin of shape (max_pos, head_dim // 2), fp32.
    Supports llama-3 style rope scaling when ``scaling`` carries
    {factor, low_freq_factor, high_freq_factor, original_max_position_embeddings}.
    """
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, dtype=torch.float64) / head_dim))
    if scaling is not None and scaling.get("rope_type", scaling.get("type")) == "llama3":
        factor = scaling["factor"]
        low = scaling["low_freq_factor"]
        high = scaling["high_freq_factor"]
        orig = scaling["original_max_position_embeddings"]
        wavelen = 2 * math.pi / inv_freq
        ratio = orig / wavelen
        smooth = ((ratio - low) / (high - low)).clamp(0.0, 1.0)
        scaled = inv_freq / factor
        smoothed = (1 - smooth) * scaled + smooth * inv_freq
        inv_freq = torch.where(wavelen > 2 * math.pi * orig / low, scaled, smoothed)
        inv_freq = torch.where(wavelen < 2 * math.pi * orig / high, 1.0 * inv_freq, inv_freq)
    t = torch.arange(max_pos, dtype=torch.float64)
    freqs = torch.outer(t, inv_freq)
    cos = freqs.cos().to(dtype)
    sin = freqs.sin().to(dtype)
    if device is not None:
        cos, sin = cos.to(device), sin.to(device)
    return cos, sin


def rope_apply(
    q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
    position_ids: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Rotary embedding with arbitrary position_ids (tree positions for
    spec-decode ride through here; ref pytorch_backend.py:59-110).

    q: (B, Hq, T, D); k: (B, Hkv, T, D); position_ids: (B, T) int64/int32.
    cos/sin: (max_pos, D/2) fp32. Rotation convention: interleaved-free
    "half-split" (HF llama): x1 = x[..., :D/2], x2 = x[..., D/2:].
    """
    B, _, T, D = q.shape
    pos = position_ids.long()
    c = cos[pos].unsqueeze(1).float()  # (B, 1, T, D/2)
    s = sin[pos].unsqueeze(1).float()

    def rot(x):
        xf = x.float()
        x1, x2 = xf[..., : D // 2], xf[..., D // 2:]
        o1 = x1 * c - x2 * s
        o2 = x2 * c + x1 * s
        return torch.cat([o1, o2], dim=-1).to(x.dtype)

    return rot(q), rot(k)


# ---------------------------------------------------------------------------
# Activation / MLP core
# ---------------------------------------------------------------------------

def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """SiLU(gate) * up (ref mlp_llama, pytorch_backend.py:1033-1048)."""
    g = gate.float()
    return (g * torch.sigmoid(g) * up.float()).to(gate.dtype)


def gelu_tanh(x: torch.Tensor) -> torch.Tensor:
    xf = x.float()
    return (0.5 * xf * (1.0 + torch.tanh(0.7978845608028654 * (xf + 0.044715 * xf ** 3)))).to(x.dtype)


# ---------------------------------------------------------------------------
# Paged KV cache primitives
# ---------------------------------------------------------------------------

def kv_write(
    k_new: torch.Tensor, v_new: torch.Tensor,
    k_pages: torch.Tensor, v_pages: torch.Tensor,
    page_table: torch.Tensor, start_pos: torch.Tensor,
) -> None:
    """Scatter new K/V into the paged cache, in place.

    k_new/v_new: (B, Hkv, T, D); k_pages: (n_pages, Hkv, P, D) token-major;
    v_pages: (n_pages, Hkv, D, P) d-major (transposed for direct MFMA
    B-fragment loads in the decode kernel — see PagedKVCache.v_pages);
    page_table: (B, max_pages) int32; start_pos: (B,) int32 — absolute position
    of k_new[:, :, 0]. (ref paged_kv.py:137-204 `write`, MI-native layout.)
    """
    B, Hkv, T, D = k_new.shape
    P = k_pages.shape[2]
    for b in range(B):
        s = int(start_pos[b])
        for t in range(T):
            pos = s + t
            page = int(page_table[b, pos // P])
            slot = pos % P
            k_pages[page, :, slot, :] = k_new[b, :, t, :]
            v_pages[page, :, :, slot] = v_new[b, :, t, :]


def kv_gather(
    k_pages: torch.Tensor, v_pages: torch.Tensor,
    page_table: torch.Tensor, ctx_len: int, batch_index: int,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Gather a sequence prefix back to dense (Hkv, ctx_len, D) tensors
    (ref paged_kv.py:265-316 `gather_prefix`)."""
    P = k_pages.shape[2]
    n = (ctx_len + P - 1) // P
    pages = page_table[batch_index, :n].long()
    k = k_pages[pages].permute(1, 0, 2, 3).reshape(k_pages.shape[1], n * P, -1)[:, :ctx_len]
    # v_pages is d-major (n_pages, Hkv, D, P) -> (Hkv, n, P, D) -> (Hkv, ctx, D)
    v = v_pages[pages].permute(1, 0, 3, 2).reshape(v_pages.shape[1], n * P, -1)[:, :ctx_len]
    return k, v


# ---------------------------------------------------------------------------
# Attention over the paged cache (decode AND prefill use the same entry)
# ---------------------------------------------------------------------------

def attn_paged(
    q: torch.Tensor,
    k_pages: torch.Tensor, v_pages: torch.Tensor,
    page_table: torch.Tensor,
    q_start: torch.Tensor,
    scale: Optional[float] = None,
    tree_mask: Optional[torch.Tensor] = None,
    sliding_window: Optional[int] = None,
    alibi_slopes: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Causal attention of q against the paged KV cache.

    Query i of sequence b sits at absolute position ``q_start[b] + i`` and
    attends to cache positions <= that position (its own k/v must already be
    written to the pages — `kv_write` first, then `attn_paged`; this single
    entry replaces the reference's separate mha_llama / mha_gen_llama pair,
    pytorch_backend.py:665-916).

    q: (B, Hq, Tq, D) -> out (B, Hq, Tq, D), GQA via Hq % Hkv == 0.
    tree_mask: (B, Tq, Tq) bool over the NEW tokens (True = may attend),
    used by tree-structured speculative verify (ref backend.py:944-1047);
    positions before q_start stay fully visible.
    sliding_window: if set, query at position p attends only to
    positions > p - sliding_window (gemma-family per-layer sliding attention).
    """
    B, Hq, Tq, D = q.shape
    Hkv = k_pages.shape[1]
    G = Hq // Hkv
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for b in range(B):
        s = int(q_start[b])
        ctx = s + Tq
        k, v = kv_gather(k_pages, v_pages, page_table, ctx, b)  # (Hkv, ctx, D)
        kf, vf = k.float(), v.float()
        qf = q[b].float()  # (Hq, Tq, D)
        # scores: (Hq, Tq, ctx)
        kg = kf.repeat_interleave(G, dim=0)
        vg = vf.repeat_interleave(G, dim=0)
        scores = torch.einsum("htd,hcd->htc", qf, kg) * scale
        if alibi_slopes is not None:
            # ALiBi (bloom family): softmax-shift-invariant absolute form
            # bias[h, :, j] = slope_h * j (HF bloom build_alibi_tensor)
            scores = scores + alibi_slopes.float().to(q.device).view(-1, 1, 1) * \
                torch.arange(ctx, dtype=torch.float32,
                             device=q.device).view(1, 1, ctx)
        dev = q.device
        pos_q = torch.arange(s, s + Tq, device=dev).unsqueeze(1)  # (Tq, 1)
        pos_k = torch.arange(ctx, device=dev).unsqueeze(0)        # (1, ctx)
        mask = pos_k <= pos_q
        if sliding_window is not None:
            mask &= pos_k > (pos_q - sliding_window)
        if tree_mask is not None:
            # new-token block (positions >= s) follows the tree mask instead
            new_block = torch.zeros(Tq, ctx, dtype=torch.bool, device=dev)
            new_block[:, s:] = tree_mask[b].to(dev)
            old_block = mask.clone()
            old_block[:, s:] = False
            mask = old_block | new_block
        scores = scores.masked_fill(~mask.to(scores.device).unsqueeze(0), float("-inf"))
        p = torch.softmax(scores, dim=-1)
        out[b] = torch.einsum("htc,hcd->htd", p, vg).to(q.dtype)
    return out


# ---------------------------------------------------------------------------
# 4-bit group quantization (ref flexgen_utils/compression.py:94-210)
# ---------------------------------------------------------------------------

def quant4_pack(x: torch.Tensor, group_size: int = 64) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Group-wise 4-bit min/max quantization along the last dim.

    Returns (packed uint8 [.., G, group_size//2], scale fp16 [.., G], zero fp16 [.., G]).
    """
    orig_shape = x.shape
    assert orig_shape[-1] % group_size == 0
    g = x.float().reshape(*orig_shape[:-1], -1, group_size)
    mn = g.min(dim=-1, keepdim=True).values
    mx = g.max(dim=-1, keepdim=True).values
    scale = (mx - mn).clamp_min(1e-8) / 15.0
    q = ((g - mn) / scale).round().clamp(0, 15).to(torch.uint8)
    packed = (q[..., 0::2] | (q[..., 1::2] << 4)).contiguous()
    return packed, scale.squeeze(-1).half(), mn.squeeze(-1).half()


def quant4_unpack(
    packed: torch.Tensor, scale: torch.Tensor, zero: torch.Tensor,
    dtype: torch.dtype = torch.bfloat16,
) -> torch.Tensor:
    lo = (packed & 0xF).float()
    hi = (packed >> 4).float()
    q = torch.stack([lo, hi], dim=-1).reshape(*packed.shape[:-1], -1)
    x = q * scale.float().unsqueeze(-1) + zero.float().unsqueeze(-1)
    return x.reshape(*packed.shape[:-2], -1).to(dtype)


def alibi_slopes(n_heads: int) -> torch.Tensor:
    """HF bloom build_alibi_tensor slopes (positive, per head; the absolute
    bias slope*j is softmax-shift-equivalent to the -slope*(i-j) distance
    penalty)."""
    import math as _m

    closest = 2 ** _m.floor(_m.log2(n_heads))
    base = 2.0 ** (-(2.0 ** -(_m.log2(closest) - 3)))
    slopes = [base ** (i + 1) for i in range(closest)]
    if closest < n_heads:
        extra_base = 2.0 ** (-(2.0 ** -(_m.log2(2 * closest) - 3)))
        slopes += [extra_base ** (2 * i + 1) for i in range(n_heads - closest)]
    return torch.tensor(slopes, dtype=torch.float32)


def attn_paged_topk(
    q: torch.Tensor,
    k_pages: torch.Tensor, v_pages: torch.Tensor,
    page_table: torch.Tensor,
    ctx_lens: torch.Tensor,
    sparsity: float,
    scale: Optional[float] = None,
    alibi_slopes: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Top-k sparse decode attention (parity: reference
    pytorch_backend.py:935-968 `_sparse_attention_value` driven by
    Policy.attn_sparsity): scores are computed against the full K cache,
    then only the top ceil(sparsity*ctx) positions' V rows participate in
    the weighted sum — the V fetch (the expensive half when the cache is
    host-resident) touches a fraction of the cache. sparsity >= 1 is exact
    dense attention. q: (B, Hq, 1, D)."""
    B, Hq, Tq, D = q.shape
    assert Tq == 1, "top-k sparse path is decode-only (one query token)"
    Hkv = k_pages.shape[1]
    G = Hq // Hkv
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for b in range(B):
        ctx = int(ctx_lens[b])
        k, v = kv_gather(k_pages, v_pages, page_table, ctx, b)  # (Hkv, ctx, D)
        kg = k.float().repeat_interleave(G, dim=0)              # (Hq, ctx, D)
        vg = v.float().repeat_interleave(G, dim=0)
        qf = q[b, :, 0].float()                                 # (Hq, D)
        scores = torch.einsum("hd,hcd->hc", qf, kg) * scale     # (Hq, ctx)
        if alibi_slopes is not None:
            scores = scores + alibi_slopes.float().to(q.device).view(-1, 1) * \
                torch.arange(ctx, dtype=torch.float32, device=q.device)
        kkeep = max(1, min(ctx, math.ceil(sparsity * ctx)))
        top = scores.topk(kkeep, dim=-1)                        # (Hq, kkeep)
        w = torch.softmax(top.values, dim=-1)                   # renormalized
        vsel = torch.gather(
            vg, 1, top.indices.unsqueeze(-1).expand(-1, -1, D))  # (Hq, kk, D)
        out[b, :, 0] = torch.einsum("hk,hkd->hd", w, vsel).to(q.dtype)
    return out


def attn_paged_mixed(
    q: torch.Tensor,
    k_pages: torch.Tensor, v_pages: torch.Tensor,
    page_table: torch.Tensor,
    ctx_lens: torch.Tensor,
    host_k: torch.Tensor, host_v: torch.Tensor,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Mixed-device decode attention (parity: reference
    pytorch_backend.py:969-1014 `_mixed_device_attention` — KV split into a
    device-resident recent segment and a host-resident old segment; the host
    part is computed fp32 on CPU and the two partial softmaxes are merged).

    host_k/host_v: (B, Hkv, S_host, D) on CPU — positions [0, S_host).
    The paged pool holds positions [S_host, S_host + ctx_lens[b]).
    q: (B, Hq, 1, D) on any device. Merge is exact (log-sum-exp), so the
    result equals dense attention over the concatenated cache.
    """
    B, Hq, Tq, D = q.shape
    assert Tq == 1, "mixed-device path is decode-only"
    Hkv = k_pages.shape[1]
    G = Hq // Hkv
    S_host = host_k.shape[2]
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for b in range(B):
        qf = q[b, :, 0].float()  # (Hq, D)
        # host segment on CPU (fp32)
        qc = qf.cpu()
        kh = host_k[b].float().repeat_interleave(G, dim=0)  # (Hq, S_host, D)
        vh = host_v[b].float().repeat_interleave(G, dim=0)
        s_h = torch.einsum("hd,hcd->hc", qc, kh) * scale
        m_h = s_h.max(-1, keepdim=True).values
        p_h = torch.exp(s_h - m_h)
        l_h = p_h.sum(-1)                                   # (Hq,)
        o_h = torch.einsum("hc,hcd->hd", p_h, vh)
        # device-resident recent segment via the paged pool
        ctx = int(ctx_lens[b])
        k, v = kv_gather(k_pages, v_pages, page_table, ctx, b)
        kg = k.float().repeat_interleave(G, dim=0)
        vg = v.float().repeat_interleave(G, dim=0)
        s_d = torch.einsum("hd,hcd->hc", qf, kg) * scale
        m_d = s_d.max(-1, keepdim=True).values
        p_d = torch.exp(s_d - m_d)
        l_d = p_d.sum(-1)
        o_d = torch.einsum("hc,hcd->hd", p_d, vg)
        # exact merge of the two partial softmaxes
        m_hd, m_dd = m_h[:, 0].to(q.device), m_d[:, 0]
        o_hd, l_hd = o_h.to(q.device), l_h.to(q.device)
        m = torch.maximum(m_hd, m_dd)
        c_h = torch.exp(m_hd - m).unsqueeze(-1)
        c_d = torch.exp(m_dd - m).unsqueeze(-1)
        denom = l_hd.unsqueeze(-1) * c_h + l_d.unsqueeze(-1) * c_d
        out[b, :, 0] = ((o_hd * c_h + o_d * c_d) / denom).to(q.dtype)
    return out


def moe_gemm_grouped(A: torch.Tensor, W: torch.Tensor, off: torch.Tensor,
                     rowmap: Optional[torch.Tensor] = None,
                     scale: Optional[torch.Tensor] = None,
                     S: Optional[int] = None) -> torch.Tensor:
    """Reference for the grouped expert GEMM (ops/hip/moe_gemm.hip):
    per-slot C[s] = A[row(s)] @ W[expert(s)]^T * scale[s]; slots grouped by
    expert via the off prefix sums. fp32 accumulation like the MFMA path."""
    E, N, K = W.shape
    offs = off.tolist()
    if S is None:
        S = offs[-1]
    C = torch.zeros(S, N, dtype=A.dtype, device=A.device)
    for e in range(E):
        a, b = offs[e], offs[e + 1]
        if b <= a:
            continue
        slots = torch.arange(a, b, device=A.device)
        rows = rowmap[slots].long() if rowmap is not None else slots
        y = A[rows].float() @ W[e].float().t()
        if scale is not None:
            y = y * scale[slots].float().unsqueeze(1)
        C[slots] = y.to(A.dtype)
    return C
