"""Server-side LLaMA decoder block on the gfx950 op library.

Replaces the reference's OptimizedLlamaDecoderLayer + FLEX_LlamaAttention/MLP
stack (models/llama/block.py:149-861, models/llama/flex_llama.py:434-793) with
a flat MI355X-native block:

  * plain-GEMM projections via hipBLASLt (torch F.linear on bf16) with QKV and
    gate/up fused into single GEMMs (fewer, larger GEMMs — xGMI/HBM sizing),
  * hand-written HIP kernels for everything fused: rmsnorm(+residual), RoPE,
    paged KV write, paged decode/prefill attention, SwiGLU,
  * one paged KV pool shared across local layers (kv/paged.py), no
    ValueHolder grids, no per-layer load/store streams: weights are resident
    in 288 GB HBM by default; the offload tier is a separate module.

Two forward paths (by design, mirroring the reference's split between
iterate_rpc_inference and run_rpc_forward/backward):
  forward_inference — HIP kernels + paged KV, no autograd.
  forward_train     — differentiable eager composition (fp32 softmax) used by
                      the fine-tuning RPCs; gradients flow to inputs/prompts
                      only (server weights frozen, ref backend.py:106-109).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from bloombee_amd import ops
from bloombee_amd.kv.paged import SessionHandle
from bloombee_amd.models.base import ModelConfig


class RopeTables:
    """Shared host-precomputed cos/sin tables, materialized per device."""

    def __init__(self, config: ModelConfig):
        self.config = config
        cos, sin = ops.rope_cos_sin(
            config.head_dim, config.max_position_embeddings,
            theta=config.rope_theta, scaling=config.rope_scaling,
        )
        self._cos, self._sin = cos, sin
        self._cache = {}

    def get(self, device: torch.device):
        key = str(device)
        if key not in self._cache:
            self._cache[key] = (self._cos.to(device), self._sin.to(device))
        return self._cache[key]


class LlamaBlock(torch.nn.Module):
    def __init__(self, config: ModelConfig, layer_index: int = 0,
                 rope: Optional[RopeTables] = None):
        super().__init__()
        self.config = config
        self.layer_index = layer_index
        H = config.hidden_size
        D = config.head_dim
        Hq, Hkv = config.num_attention_heads, config.num_key_value_heads
        I = config.intermediate_size
        dt = config.dtype
        self.Hq, self.Hkv, self.D, self.I = Hq, Hkv, D, I
        self.scale = 1.0 / math.sqrt(D)

        def p(*shape):
            return torch.nn.Parameter(torch.empty(*shape, dtype=dt), requires_grad=False)

        self.input_norm_w = p(H)
        self.qkv_w = p((Hq + 2 * Hkv) * D, H)     # fused q|k|v
        self.o_w = p(H, Hq * D)
        self.post_norm_w = p(H)
        self.gate_up_w = p(2 * I, H)              # fused gate|up
        self.down_w = p(H, I)
        self.rope = rope if rope is not None else RopeTables(config)
        # fused-norm chain state (forward_inference fuse_norm path):
        # per-stripe row sum-of-squares buffers, (H/64, 32) f32, written by
        # the o/down GEMM epilogues and consumed by the next fused rmsnorm
        self._ss_h2 = None
        self._ss_hidden = None
        self._ss_valid = False

    @torch.no_grad()
    def init_random(self, seed: Optional[int] = None):
        """Random-init weights. Generates on the parameters' current device —
        device-side RNG for resident blocks (seconds for 70B-class shards vs
        minutes of host randn + transfer). Same (device kind, seed) => same
        weights, which is what the multi-rank parity argument needs."""
        s = seed if seed is not None else 1234 + self.layer_index
        dev = self.input_norm_w.device
        gen = torch.Generator(device=dev).manual_seed(s)
        std = 0.02 / math.sqrt(2 * self.config.num_hidden_layers)
        for name, w in self.named_parameters():
            if name.endswith("norm_w"):
                w.fill_(1.0)
            else:
                w.copy_(torch.randn(w.shape, generator=gen, dtype=torch.float32,
                                    device=dev).mul_(std).to(w.dtype))
        return self

    def _fuse_norm_gate(self, hidden) -> bool:
        """All four decode GEMMs must hit the v2 kernel (K%256) with
        stripe-aligned N so the fused-norm ss chain stays on-device;
        W4/LoRA blocks keep the separate-norm path."""
        return (getattr(self, "_w4", None) is None
                and getattr(self, "lora_delta", None) is None
                and ops.fuse_norm_linear_ok(hidden, self.qkv_w)
                and self.I % 256 == 0 and (2 * self.I) % 64 == 0
                and (self.Hq * self.D) % 256 == 0
                and self.config.hidden_size % 64 == 0)

    def _ensure_ss_bufs(self, device) -> None:
        if self._ss_h2 is None or self._ss_h2.device != device:
            st = self.config.hidden_size // 64
            self._ss_h2 = torch.empty(st * 32, dtype=torch.float32,
                                      device=device)
            self._ss_hidden = torch.empty_like(self._ss_h2)

    @torch.no_grad()
    def fold_norm_weights(self):
        """Reparameterize for the fused-norm decode path: qkv_w <- qkv_w *
        input_norm_w, gate_up_w <- gate_up_w * post_norm_w (f32 product,
        one bf16 round), norm weights <- 1. rms_norm with unit weight is
        mathematically unchanged on EVERY path (train, prefill, CPU), and
        the decode GEMM then only applies the per-row 1/rms in its
        epilogue (norm_mode 2) instead of per-element weight scaling
        (profiles/r02 §13: the per-stage scale cost +4-8 us/GEMM)."""
        if getattr(self, "_norm_folded", False):
            return
        self._norm_fold_orig = (self.input_norm_w.detach().clone(),
                                self.post_norm_w.detach().clone())
        self.qkv_w.data = (self.qkv_w.float()
                           * self.input_norm_w.float()).to(self.qkv_w.dtype)
        self.gate_up_w.data = (self.gate_up_w.float()
                               * self.post_norm_w.float()).to(self.gate_up_w.dtype)
        self.input_norm_w.data.fill_(1.0)
        self.post_norm_w.data.fill_(1.0)
        self._norm_folded = True

    @torch.no_grad()
    def unfold_norm_weights(self):
        """Best-effort inverse of fold_norm_weights (division re-rounds:
        ~1 ulp drift on the projections). Needed before attaching LoRA
        adapters trained against the original parameterization."""
        if not getattr(self, "_norm_folded", False):
            return
        inw, pnw = self._norm_fold_orig
        self.input_norm_w.data.copy_(inw)
        self.post_norm_w.data.copy_(pnw)
        den_i = inw.float().abs().clamp_min(1e-6) * inw.float().sign().where(
            inw.float() != 0, torch.ones_like(inw.float()))
        den_p = pnw.float().abs().clamp_min(1e-6) * pnw.float().sign().where(
            pnw.float() != 0, torch.ones_like(pnw.float()))
        self.qkv_w.data = (self.qkv_w.float() / den_i).to(self.qkv_w.dtype)
        self.gate_up_w.data = (self.gate_up_w.float() / den_p).to(self.gate_up_w.dtype)
        self._norm_folded = False

    @torch.no_grad()
    def quantize_weights_q4(self, keep_full: bool = False):
        """Pack the four projection weights into 4-bit group codes
        (quant4_pack layout); decode GEMMs then stream ~3.5x fewer bytes
        through w4_gemm.hip (FlexGen-style compression ON the compute
        path — the reference only compresses weights at rest,
        flexgen_utils/compression.py:94-210). keep_full=False drops the
        bf16 weights (4x memory; the block becomes inference-only and
        prefill-shaped GEMMs dequantize on the fly)."""
        self._w4 = {}
        for name in ("qkv_w", "o_w", "gate_up_w", "down_w"):
            w = getattr(self, name).detach()
            packed, scale, zero = ops.quant4_pack(w)
            N = w.shape[0]
            self._w4[name] = (packed.reshape(N, -1).contiguous(),
                              scale.reshape(N, -1).half().contiguous(),
                              zero.reshape(N, -1).half().contiguous(), N)
            if not keep_full:
                getattr(self, name).data = torch.empty(
                    0, dtype=w.dtype, device=w.device)
        return self

    def _lin(self, x, w, name, **kw):
        """ops.linear + optional active-LoRA delta (utils/peft.py).
        Quantized blocks (quantize_weights_q4) route through the 4-bit
        weight-stream kernel instead."""
        w4 = getattr(self, "_w4", None)
        if w4 is not None and name in w4:
            packed, scale, zero, N = w4[name]
            y = ops.linear_w4(x, packed, scale, zero, N, **kw)
        else:
            y = ops.linear(x, w, **kw)
        ld = getattr(self, "lora_delta", None)
        if ld is not None:
            d = ld(name, x)
            if d is not None:
                y = y + d
        return y

    # ------------------------------------------------------------------
    # inference path (HIP kernels, paged KV)
    # ------------------------------------------------------------------
    @torch.no_grad()
    def forward_inference(
        self,
        hidden: torch.Tensor,              # (B, T, H) bf16
        kv: SessionHandle,
        start_pos: torch.Tensor,           # (B,) int32 — absolute pos of token 0
        position_ids: Optional[torch.Tensor] = None,  # (B, T) int32
        tree_mask: Optional[torch.Tensor] = None,     # (B, T, T) bool, spec verify
    ) -> torch.Tensor:
        B, T, H = hidden.shape
        Hq, Hkv, D = self.Hq, self.Hkv, self.D
        cfg = self.config

        # Decode-shaped GPU steps fold both per-layer rmsnorms into the
        # skinny GEMM's A-operand stage (ops.linear norm=...): two fewer
        # launches + elementwise passes per layer. The row sum-of-squares
        # the norm needs is produced by the PRECEDING GEMM's epilogue
        # (ss buffers chained block-to-block via prev_block; re-streaming
        # A instead costs +18-60 us/GEMM — profiles/r02 §13). W4/LoRA
        # blocks and prefill keep the separate-norm path.
        fuse_norm = self._fuse_norm_gate(hidden)
        self._ss_valid = False
        if fuse_norm and not getattr(self, "_norm_folded", False):
            self.fold_norm_weights()   # one-time reparameterization
        pb = getattr(self, "prev_block", None)
        if fuse_norm and pb is not None and getattr(pb, "_ss_valid", False):
            # previous block's down-proj left sum-of-squares for `hidden`;
            # norm weight is folded into qkv_w -> invr-only norm (mode 2)
            qkv = ops.linear(hidden, self.qkv_w,
                             norm=(None, cfg.rms_norm_eps),
                             ss_in=pb._ss_hidden)
        else:
            x = ops.rms_norm(hidden, self.input_norm_w, cfg.rms_norm_eps)
            qkv = self._lin(x, self.qkv_w, "qkv_w")        # (B, T, (Hq+2Hkv)D)
        cos, sin = self.rope.get(hidden.device)
        kp = kv.k_pages(self.layer_index)
        vp = kv.v_pages(self.layer_index)
        pt = kv.page_table()
        # mixed-device sessions (host KV prefix + device recent segment,
        # kv.swap_in_as_prefix): paged indices are LOCAL to the device pool
        # but rotary positions must stay ABSOLUTE
        off = getattr(kv, "pos_offset", 0)
        if off and position_ids is None:
            position_ids = (start_pos.view(B, 1) + off
                            + torch.arange(T, device=hidden.device)
                            .view(1, T)).int()
        # fused RoPE + paged KV write on the raw GEMM output, then attention
        # reads q in place and emits the O-projection input — no transposes
        ops.rope_kv_write_(qkv, Hq, Hkv, cos, sin, position_ids, kp, vp, pt,
                           start_pos)
        hp = (kv.host_prefix(self.layer_index)
              if hasattr(kv, "host_prefix") else None)
        if hp is not None:
            # exact two-segment merge (ref _mixed_device_attention): the
            # host prefix is attended on the CPU, the recent segment via
            # the paged pool; decode-only (backend gates T == 1)
            D_ = self.D
            q = (qkv[..., : Hq * D_].view(B, T, Hq, D_)
                 .permute(0, 2, 1, 3).contiguous())
            attn = ops.attn_paged_mixed(q, kp, vp, pt, start_pos + T,
                                        hp[0], hp[1], self.scale)
            attn = attn.permute(0, 2, 1, 3).reshape(B, T, Hq * D_)
        elif tree_mask is not None:
            # tree-structured verify step (spec decoding): unfused q view +
            # the tree-mask attention path
            q = (qkv[..., : Hq * D].view(B, T, Hq, D)
                 .permute(0, 2, 1, 3).contiguous())
            attn = ops.attn_paged(q, kp, vp, pt, start_pos.long(), self.scale,
                                  tree_mask=tree_mask)
            attn = attn.permute(0, 2, 1, 3).reshape(B, T, Hq * D)
        else:
            attn = ops.attn_paged_qkv(qkv, Hq, Hkv, kp, vp, pt, start_pos,
                                      self.scale)
        if fuse_norm:
            # h2 = hidden + o(attn) via the GEMM's residual epilogue, which
            # also emits row sum-of-squares; the post-attention rmsnorm
            # folds into the gate_up A-stage consuming them. down-proj
            # leaves the stats for the NEXT block's input norm.
            self._ensure_ss_bufs(hidden.device)
            h2 = ops.linear(attn, self.o_w, residual=hidden,
                            ss_out=self._ss_h2)
            gu = ops.linear(h2, self.gate_up_w,
                            norm=(None, cfg.rms_norm_eps),
                            ss_in=self._ss_h2)
            out = ops.linear(ops.swiglu(gu), self.down_w, residual=h2,
                             ss_out=self._ss_hidden)
            self._ss_valid = True
            return out
        a = self._lin(attn, self.o_w, "o_w")

        # h2 = hidden + a fused into the post-attention norm
        h2, y = ops.rms_norm_residual(a, hidden, self.post_norm_w, cfg.rms_norm_eps)
        # residual add fused into the down-projection epilogue
        return self._lin(ops.swiglu(self._lin(y, self.gate_up_w, "gate_up_w")),
                         self.down_w, "down_w", residual=h2)

    # ------------------------------------------------------------------
    # training path (differentiable; full sequence, no KV cache)
    # ------------------------------------------------------------------
    def forward_train(self, hidden: torch.Tensor, start_pos: int = 0) -> torch.Tensor:
        B, T, H = hidden.shape
        Hq, Hkv, D = self.Hq, self.Hkv, self.D
        cfg = self.config
        G = Hq // Hkv

        def rms(x, w):
            xf = x.float()
            return (xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + cfg.rms_norm_eps)
                    ).to(x.dtype) * w

        def lin(x_, w, name):
            # training path honors active LoRA adapters too (the inference
            # path routes through self._lin; keep both consistent)
            y = F.linear(x_, w)
            ld = getattr(self, "lora_delta", None)
            if ld is not None:
                d = ld(name, x_)
                if d is not None:
                    y = y + d
            return y

        x = rms(hidden, self.input_norm_w)
        qkv = lin(x, self.qkv_w, "qkv_w").view(B, T, Hq + 2 * Hkv, D).permute(0, 2, 1, 3)
        q, k, v = qkv.split([Hq, Hkv, Hkv], dim=1)
        cos, sin = self.rope.get(hidden.device)
        pos = torch.arange(start_pos, start_pos + T, device=hidden.device)
        c = cos[pos].view(1, 1, T, D // 2).float()
        s = sin[pos].view(1, 1, T, D // 2).float()

        def rot(t):
            tf = t.float()
            t1, t2 = tf[..., : D // 2], tf[..., D // 2:]
            return torch.cat([t1 * c - t2 * s, t2 * c + t1 * s], -1).to(t.dtype)

        q, k = rot(q), rot(k)
        k = k.repeat_interleave(G, dim=1)
        v = v.repeat_interleave(G, dim=1)
        scores = torch.matmul(q.float(), k.float().transpose(-1, -2)) * self.scale
        mask = torch.ones(T, T, dtype=torch.bool, device=hidden.device).tril()
        scores = scores.masked_fill(~mask, float("-inf"))
        p = torch.softmax(scores, dim=-1)
        attn = torch.matmul(p, v.float()).to(hidden.dtype)
        attn = attn.permute(0, 2, 1, 3).reshape(B, T, Hq * D)
        h2 = hidden + lin(attn, self.o_w, "o_w")
        y = rms(h2, self.post_norm_w)
        gu = F.linear(y, self.gate_up_w)
        g, u = gu.split([self.I, self.I], dim=-1)
        m = F.linear(F.silu(g.float()).to(u.dtype) * u, self.down_w)
        return h2 + m

    def forward(self, *args, **kw):
        return self.forward_inference(*args, **kw)
