"""Dispatch layer between the gfx950 HIP extension and the torch reference.

CPU tensors run the fp32 reference (bloombee_amd/ops/reference.py); device
tensors run the in-tree _hip_ops extension and FAIL LOUDLY if it is absent —
a silent eager fallback on a GPU box would invalidate every GPU test
(the round-end driver records which .so the GPU processes actually load).
"""
from __future__ import annotations

import importlib
import math
import os
from typing import Optional, Tuple

import torch

from bloombee_amd.ops import reference as ref
from bloombee_amd.ops.reference import alibi_slopes as alibi_slopes_for  # noqa: F401
from bloombee_amd.ops.reference import rope_cos_sin  # noqa: F401  (host-side)
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)

try:
    hip_ops = importlib.import_module("bloombee_amd.ops._hip_ops")
    HAVE_HIP_OPS = True
except ImportError as e:  # pragma: no cover
    hip_ops = None
    HAVE_HIP_OPS = False
    _import_error = e


def _require_ext():
    if not HAVE_HIP_OPS:
        raise RuntimeError(
            "bloombee_amd HIP extension is not built but a tensor is on GPU. "
            "Run `python setup.py hip` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error was: {_import_error}"
        )


def _on_gpu(t: torch.Tensor) -> bool:
    return t.is_cuda


# ---------------------------------------------------------------------------


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if _on_gpu(x):
        _require_ext()
        return hip_ops.rms_norm(x.contiguous(), weight.contiguous(), eps)
    return ref.rms_norm(x, weight, eps)


def rms_norm_residual(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
) -> Tuple[torch.Tensor, torch.Tensor]:
    if _on_gpu(x):
        _require_ext()
        h, y = hip_ops.rms_norm_residual(x.contiguous(), residual.contiguous(),
                                         weight.contiguous(), eps)
        return h, y
    return ref.rms_norm_residual(x, residual, weight, eps)


def layer_norm(x, weight, bias=None, eps: float = 1e-5):
    if _on_gpu(x):
        _require_ext()
        return hip_ops.layer_norm(x.contiguous(), weight.contiguous(),
                                  bias.contiguous() if bias is not None else None, eps)
    return ref.layer_norm(x, weight, bias, eps)


def rope_apply_(q, k, cos, sin, position_ids):
    """In-place RoPE on q, k: (B, H*, T, D). position_ids: (B, T) int32."""
    if _on_gpu(q):
        _require_ext()
        hip_ops.rope_apply_(q, k, cos, sin, position_ids.int())
        return q, k
    q2, k2 = ref.rope_apply(q, k, cos, sin, position_ids)
    q.copy_(q2)
    k.copy_(k2)
    return q, k


def swiglu(gate_up: torch.Tensor) -> torch.Tensor:
    """gate_up: (..., 2I) from the fused gate/up GEMM -> (..., I)."""
    if _on_gpu(gate_up):
        _require_ext()
        return hip_ops.swiglu(gate_up.contiguous())
    I = gate_up.shape[-1] // 2
    return ref.swiglu(gate_up[..., :I], gate_up[..., I:])


def gelu_tanh(x: torch.Tensor) -> torch.Tensor:
    if _on_gpu(x):
        _require_ext()
        return hip_ops.gelu_tanh(x.contiguous())
    return ref.gelu_tanh(x)


def kv_write(k_new, v_new, k_pages, v_pages, page_table, start_pos):
    if _on_gpu(k_new):
        _require_ext()
        hip_ops.kv_write(k_new.contiguous(), v_new.contiguous(), k_pages, v_pages,
                         page_table, start_pos.int())
        return
    ref.kv_write(k_new, v_new, k_pages, v_pages, page_table, start_pos)


def kv_gather(k_pages, v_pages, page_table, ctx_len: int, batch_index: int):
    if _on_gpu(k_pages):
        _require_ext()
        k, v = hip_ops.kv_gather(k_pages, v_pages, page_table, batch_index, ctx_len)
        return k, v
    return ref.kv_gather(k_pages, v_pages, page_table, ctx_len, batch_index)


_attn_sparsity = 1.0  # Policy.attn_sparsity (ref flexgen policy :10-55)
# fused tree-mask attention: default ON since round 2 (numerics validated
# on MI355X — tests/test_treekernel.py 3/3 after the bool-dispatch fix);
# BBAMD_TREE_KERNEL=0 falls back to the torch composition
_TREE_KERNEL = os.environ.get("BBAMD_TREE_KERNEL", "1") not in ("", "0")


def set_attn_sparsity(frac: float) -> None:
    """Enable top-k sparse decode attention globally (parity: reference
    Policy.attn_sparsity — a server-wide policy knob, not per-call). frac
    is the fraction of cache positions whose V rows participate; 1.0
    restores exact dense attention."""
    global _attn_sparsity
    _attn_sparsity = float(frac)


def attn_decode(
    q, k_pages, v_pages, page_table, ctx_lens, scale: Optional[float] = None,
    window: int = 0, n_split: int = 0, alibi_slopes=None,
) -> torch.Tensor:
    """Single-token paged attention. q: (B, Hq, 1, D)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _attn_sparsity < 1.0 and window <= 0:
        return ref.attn_paged_topk(q, k_pages, v_pages, page_table, ctx_lens,
                                   _attn_sparsity, scale,
                                   alibi_slopes=alibi_slopes)
    if _on_gpu(q):
        _require_ext()
        if q.shape[-1] > 256 and q.shape[-1] != 512:
            # odd wide head dims: gather-based torch composition on-device
            # (D=512 runs attn_decode_wide_kernel — d split across waves)
            return ref.attn_paged(q, k_pages, v_pages, page_table,
                                  ctx_lens.long() - 1, scale,
                                  sliding_window=window if window > 0 else None,
                                  alibi_slopes=alibi_slopes)
        al = (alibi_slopes.float().to(q.device).contiguous()
              if alibi_slopes is not None else None)
        return hip_ops.attn_decode(q.contiguous(), k_pages, v_pages, page_table,
                                   ctx_lens.int(), scale, window, n_split, al)
    q_start = ctx_lens.long() - 1
    return ref.attn_paged(q, k_pages, v_pages, page_table, q_start, scale,
                          sliding_window=window if window > 0 else None,
                          alibi_slopes=alibi_slopes)


def attn_prefill(
    q, k_pages, v_pages, page_table, q_start, scale: Optional[float] = None,
    window: int = 0, alibi_slopes=None,
) -> torch.Tensor:
    """Multi-token causal paged attention. q: (B, Hq, Tq, D); the new tokens'
    K/V must already be in the pages (kv_write first)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _on_gpu(q):
        _require_ext()
        if q.shape[-1] > 256:
            return ref.attn_paged(q, k_pages, v_pages, page_table,
                                  q_start.long(), scale,
                                  sliding_window=window if window > 0 else None,
                                  alibi_slopes=alibi_slopes)
        al = (alibi_slopes.float().to(q.device).contiguous()
              if alibi_slopes is not None else None)
        return hip_ops.attn_prefill(q.contiguous(), k_pages, v_pages, page_table,
                                    q_start.int(), scale, window, al, None)
    return ref.attn_paged(q, k_pages, v_pages, page_table, q_start, scale,
                          sliding_window=window if window > 0 else None,
                          alibi_slopes=alibi_slopes)


def rope_kv_write_(qkv, Hq: int, Hkv: int, cos, sin, position_ids,
                   k_pages, v_pages, page_table, start_pos):
    """Fused RoPE + paged KV write on the raw QKV GEMM output.

    qkv: (B, T, (Hq+2Hkv)*D) — q section roped in place; k roped -> pages;
    v copied -> pages. position_ids None => start_pos[b] + t.
    """
    if _on_gpu(qkv):
        _require_ext()
        hip_ops.rope_kv_write_(qkv, Hq, Hkv, cos, sin,
                               position_ids.int() if position_ids is not None else None,
                               k_pages, v_pages, page_table, start_pos.int())
        return
    B, T, _ = qkv.shape
    D = k_pages.shape[3]
    q = qkv[..., : Hq * D].view(B, T, Hq, D).permute(0, 2, 1, 3)
    k = qkv[..., Hq * D:(Hq + Hkv) * D].view(B, T, Hkv, D).permute(0, 2, 1, 3)
    v = qkv[..., (Hq + Hkv) * D:].view(B, T, Hkv, D).permute(0, 2, 1, 3)
    if position_ids is None:
        position_ids = start_pos.view(B, 1).long() + torch.arange(T).view(1, T)
    q2, k2 = ref.rope_apply(q, k, cos, sin, position_ids)
    qkv[..., : Hq * D].copy_(q2.permute(0, 2, 1, 3).reshape(B, T, Hq * D))
    ref.kv_write(k2, v.contiguous(), k_pages, v_pages, page_table, start_pos)


def attn_paged_qkv(qkv, Hq: int, Hkv: int, k_pages, v_pages, page_table,
                   q_start, scale=None, window: int = 0):
    """Attention reading q straight from the fused QKV buffer.

    qkv: (B, T, (Hq+2Hkv)*D) with the q section already roped and the new
    tokens' K/V already in the pages (rope_kv_write_ first).
    Returns (B, T, Hq*D) — the O-projection GEMM input, no transposes.
    """
    B, T, _ = qkv.shape
    D = k_pages.shape[3]
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    if _on_gpu(qkv):
        _require_ext()
        if T == 1:
            ctx = q_start + 1
            return hip_ops.attn_decode_qkv(qkv, Hq, k_pages, v_pages, page_table,
                                           ctx.int(), scale, window, 0)
        return hip_ops.attn_prefill_qkv(qkv, Hq, k_pages, v_pages, page_table,
                                        q_start.int(), scale, window)
    q = qkv[..., : Hq * D].view(B, T, Hq, D).permute(0, 2, 1, 3).contiguous()
    out = ref.attn_paged(q, k_pages, v_pages, page_table, q_start.long(), scale,
                         sliding_window=window if window > 0 else None)
    return out.permute(0, 2, 1, 3).reshape(B, T, Hq * D)


def attn_paged(q, k_pages, v_pages, page_table, q_start, scale=None, window: int = 0,
               tree_mask=None, alibi_slopes=None):
    """Unified entry: picks decode vs prefill kernel by Tq."""
    Tq = q.shape[2]
    if alibi_slopes is not None and not _on_gpu(q):
        if scale is None:
            scale = 1.0 / math.sqrt(q.shape[-1])
        return ref.attn_paged(q, k_pages, v_pages, page_table, q_start, scale,
                              alibi_slopes=alibi_slopes,
                              sliding_window=window if window > 0 else None)
    if tree_mask is not None:
        if scale is None:
            scale = 1.0 / math.sqrt(q.shape[-1])
        if (_on_gpu(q) and _TREE_KERNEL and window == 0
                and q.shape[-1] <= 256 and alibi_slopes is None):
            # fused tree-mask prefill kernel (spec verify) — the default
            # GPU path (tests/test_treekernel.py parity on MI355X)
            _require_ext()
            tm = tree_mask.to(q.device).contiguous().bool()
            return hip_ops.attn_prefill(q.contiguous(), k_pages, v_pages,
                                        page_table, q_start.int(), scale, 0,
                                        None, tm)
        # tree-attention (spec decode verify): device-agnostic torch
        # composition (ref.attn_paged runs on GPU tensors directly)
        return ref.attn_paged(q, k_pages, v_pages, page_table,
                              q_start.to(q.device), scale,
                              tree_mask=tree_mask.to(q.device))
    if Tq == 1:
        ctx_lens = q_start + 1
        return attn_decode(q, k_pages, v_pages, page_table, ctx_lens, scale,
                           window, alibi_slopes=alibi_slopes)
    return attn_prefill(q, k_pages, v_pages, page_table, q_start, scale, window,
                        alibi_slopes=alibi_slopes)


def linear(x: torch.Tensor, w: torch.Tensor, residual: Optional[torch.Tensor] = None,
           bias: Optional[torch.Tensor] = None,
           norm: Optional[tuple] = None,
           ss_in: Optional[torch.Tensor] = None,
           ss_out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """y = (rmsnorm(x) if norm else x) @ w^T (+bias) (+residual).
    Decode-shaped GEMMs (<=32 rows) on GPU go through the skinny-M MFMA
    kernel (gemm_skinny.hip) — hipBLASLt leaves ~2x weight-stream bandwidth
    on the table at M<=32; everything else uses hipBLASLt via F.linear.
    norm=(weight, eps) folds the input rmsnorm into the kernel's A-operand
    stage (launch-count reduction; the separate rms_norm launch + its
    read/write disappear from the decode step). ss_in is the producing
    GEMM's per-stripe row sum-of-squares (its ss_out) so the fused norm
    skips re-streaming A; ss_out asks this GEMM's epilogue to emit the
    same stats for ITS consumer (f32 (N/64, 32), overwritten — graph-safe)."""
    M = x.numel() // x.shape[-1]
    N, K = w.shape[0], x.shape[-1]
    # All decode-shaped GEMMs route to the skinny kernel: although hipBLASLt
    # wins the two MLP shapes in an isolated microbench, in the real decode
    # step it runs ~1.5x slower (cold L2 / per-stream heuristics, see
    # profiles/), while gemm_skinny runs faster in-context AND fuses the
    # residual/bias epilogue the blaslt path would re-emit as a separate
    # eager kernel.
    if (_on_gpu(x) and M <= 32 and x.dtype == torch.bfloat16
            and K % 256 == 0 and N % 64 == 0):
        _require_ext()
        if norm is not None or ss_out is not None:
            nw = None
            eps = 0.0
            mode = 0
            if norm is not None:
                eps = float(norm[1])
                if norm[0] is None:
                    mode = 2  # weight pre-folded into W: invr-only scaling
                else:
                    nw = norm[0].contiguous()
                    mode = 1
            return hip_ops.gemm_skinny(x.contiguous(), w, residual, bias, 0,
                                       nw, eps, ss_in, ss_out, mode)
        return hip_ops.gemm_skinny(x.contiguous(), w, residual, bias, 0)
    if norm is not None:
        nw = norm[0]
        if nw is None:
            nw = torch.ones(x.shape[-1], dtype=x.dtype, device=x.device)
        x = rms_norm(x, nw, norm[1])
    y = torch.nn.functional.linear(x, w, bias)
    if residual is not None:
        y = y + residual.view_as(y)
    if ss_out is not None:
        raise RuntimeError("ss_out requires the GPU skinny-GEMM path "
                           "(callers gate on fuse_norm_linear_ok)")
    return y


_FUSE_NORM = os.environ.get("BBAMD_FUSE_NORM", "1") != "0"


def fuse_norm_linear_ok(x: torch.Tensor, w: torch.Tensor) -> bool:
    """True when linear(norm=...) will take the fused-kernel path (so the
    caller can skip emitting a separate rms_norm). Default ON
    (BBAMD_FUSE_NORM=0 reverts): +2.9% on the flagship decode bench
    (5289 vs 5140 tok/s) once the row stats ride the producer GEMM's
    epilogue and the norm weights are folded into the projections
    (profiles/r02 §13; the naive per-WG A-restream variant measured 16%
    SLOWER — the chain is what makes it pay)."""
    K = x.shape[-1]
    return (_FUSE_NORM and _on_gpu(x) and x.numel() // K <= 32
            and x.dtype == torch.bfloat16 and K % 256 == 0
            and w.shape[0] % 64 == 0 and HAVE_HIP_OPS)


def attn_paged_mixed(q, k_pages, v_pages, page_table, ctx_lens,
                     host_k, host_v, scale=None):
    """Mixed-device decode attention: host-resident KV prefix (CPU fp32)
    + device-resident recent segment, merged with the exact log-sum-exp
    composition (ref _mixed_device_attention, pytorch_backend.py:969-1014).
    Device-agnostic torch composition — the host segment's matmuls RUN on
    the CPU by construction, which is the point of the mode."""
    return ref.attn_paged_mixed(q, k_pages, v_pages, page_table, ctx_lens,
                                host_k, host_v, scale)


def moe_gemm_grouped(A: torch.Tensor, W: torch.Tensor, off: torch.Tensor,
                     rowmap: Optional[torch.Tensor] = None,
                     scale: Optional[torch.Tensor] = None,
                     S: Optional[int] = None,
                     mchunks: Optional[int] = None) -> torch.Tensor:
    """Grouped expert GEMM, one launch over all experts (moe_gemm.hip):
    C[s] = A[rowmap[s] if rowmap else s] @ W[expert(s)]^T (* scale[s]).

    Slots are expert-sorted: [off[e], off[e+1]) belong to expert e; off is
    the int32 exclusive prefix sum ON DEVICE (no host sync — hipGraph-safe,
    unlike the per-expert nonzero() loop). S = total slots, mchunks = host
    upper bound on ceil(max rows per expert / 32) (worst case: every token
    routed to one expert). Replaces the per-expert skinny GEMM loop for
    Mixtral decode (BASELINE config 4 "MoE grouped GEMM").
    """
    E, N, K = W.shape
    if S is None:
        S = int(rowmap.numel()) if rowmap is not None else int(A.shape[0])
    if mchunks is None:
        mchunks = (int(A.shape[0]) + 31) // 32
    if (_on_gpu(A) and A.dtype == torch.bfloat16 and K % 32 == 0
            and N % 64 == 0):
        _require_ext()
        return hip_ops.moe_gemm(A.contiguous(), W.contiguous(),
                                off.int(), rowmap, scale, S, mchunks)
    return ref.moe_gemm_grouped(A, W, off, rowmap, scale, S)


def linear_w4(x: torch.Tensor, packed: torch.Tensor, scale: torch.Tensor,
              zero: torch.Tensor, N: int,
              residual: Optional[torch.Tensor] = None,
              bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """y = x @ dequant4(W)^T (+bias) (+residual) with W in quant4_pack
    layout (N, K/2 codes + f16 scale/zero per 64-group). Decode shapes
    (M<=32) run the 4-bit weight-stream kernel (w4_gemm.hip, ~3.5x less
    HBM traffic than bf16); anything else dequantizes and uses the dense
    path — W4 is a decode-bandwidth play, prefill is compute-bound."""
    K = x.shape[-1]
    M = x.numel() // K
    if (_on_gpu(x) and x.dtype == torch.bfloat16 and M <= 32
            and K % 128 == 0 and N % 64 == 0):
        _require_ext()
        # the kernel computes in f16 (fast nibble->half dequant + f16 MFMA);
        # the activation conversion is M*K elements - noise next to the
        # 4-bit weight stream it unlocks
        return hip_ops.gemm_w4(x.half().contiguous(),
                               packed.reshape(N, K // 2),
                               scale.reshape(N, K // 64).half(),
                               zero.reshape(N, K // 64).half(),
                               residual, bias, N, 0)
    w = quant4_unpack(packed.reshape(-1, 32), scale.reshape(-1),
                      zero.reshape(-1), dtype=x.dtype).reshape(N, K)
    y = torch.nn.functional.linear(x, w, bias)
    if residual is not None:
        y = y + residual.view_as(y)
    return y


def quant4_pack(x: torch.Tensor, group_size: int = 64):
    if _on_gpu(x):
        _require_ext()
        assert group_size == 64
        packed, scale, zero = hip_ops.quant4_pack(x.contiguous())
        return packed, scale, zero
    return ref.quant4_pack(x, group_size)


def quant4_unpack(packed, scale, zero, dtype=torch.bfloat16):
    if _on_gpu(packed):
        _require_ext()
        return hip_ops.quant4_unpack(packed, scale, zero)
    return ref.quant4_unpack(packed, scale, zero, dtype)


def mfma_selftest(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    _require_ext()
    return hip_ops.mfma_selftest(A.contiguous(), B.contiguous())
