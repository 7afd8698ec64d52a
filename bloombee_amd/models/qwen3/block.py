"""Server-side Qwen3 block (parity: reference models/qwen3/block.py
WrappedQwen3Block :18-181): LLaMA-style GQA block plus per-head RMS q/k norms
applied on the raw QKV GEMM output before RoPE. head_dim is decoupled from
hidden_size (Qwen3-8B: D=128 with Hq*D != hidden)."""
from __future__ import annotations

import math
from typing import Optional

import torch

from bloombee_amd import ops
from bloombee_amd.kv.paged import SessionHandle
from bloombee_amd.models.base import ModelConfig
from bloombee_amd.models.llama.block import LlamaBlock, RopeTables


class Qwen3Block(LlamaBlock):
    def __init__(self, config: ModelConfig, layer_index: int = 0,
                 rope: Optional[RopeTables] = None):
        super().__init__(config, layer_index, rope)
        D = config.head_dim
        dt = config.dtype
        self.q_norm_w = torch.nn.Parameter(torch.empty(D, dtype=dt),
                                           requires_grad=False)
        self.k_norm_w = torch.nn.Parameter(torch.empty(D, dtype=dt),
                                           requires_grad=False)

    @torch.no_grad()
    def forward_inference(self, hidden: torch.Tensor, kv: SessionHandle,
                          start_pos: torch.Tensor,
                          position_ids=None) -> torch.Tensor:
        B, T, H = hidden.shape
        Hq, Hkv, D = self.Hq, self.Hkv, self.D
        cfg = self.config

        # fused-norm chain (see llama/block.py): both rmsnorms ride the
        # GEMMs; the per-head q/k norms below act on the GEMM OUTPUT and
        # are unaffected by the input-norm weight fold
        fuse_norm = self._fuse_norm_gate(hidden)
        self._ss_valid = False
        if fuse_norm and not getattr(self, "_norm_folded", False):
            self.fold_norm_weights()
        pb = getattr(self, "prev_block", None)
        if fuse_norm and pb is not None and getattr(pb, "_ss_valid", False):
            qkv = ops.linear(hidden, self.qkv_w,
                             norm=(None, cfg.rms_norm_eps),
                             ss_in=pb._ss_hidden)
        else:
            x = ops.rms_norm(hidden, self.input_norm_w, cfg.rms_norm_eps)
            qkv = self._lin(x, self.qkv_w, "qkv_w")
        # per-head q/k RMS norm on the fused buffer (contiguous D-sized rows)
        qk = qkv[..., :(Hq + Hkv) * D]
        q_flat = qkv[..., :Hq * D].reshape(-1, D)
        k_flat = qkv[..., Hq * D:(Hq + Hkv) * D].reshape(-1, D)
        qkv[..., :Hq * D] = ops.rms_norm(q_flat, self.q_norm_w,
                                         cfg.rms_norm_eps).view(B, T, Hq * D)
        qkv[..., Hq * D:(Hq + Hkv) * D] = ops.rms_norm(
            k_flat, self.k_norm_w, cfg.rms_norm_eps).view(B, T, Hkv * D)
        cos, sin = self.rope.get(hidden.device)
        kp = kv.k_pages(self.layer_index)
        vp = kv.v_pages(self.layer_index)
        pt = kv.page_table()
        # mixed-device sessions (host KV prefix + device recent segment):
        # pool positions are local, rotary stays absolute (see llama block)
        off = getattr(kv, "pos_offset", 0)
        if off and position_ids is None:
            position_ids = (start_pos.view(B, 1) + off
                            + torch.arange(T, device=hidden.device)
                            .view(1, T)).int()
        ops.rope_kv_write_(qkv, Hq, Hkv, cos, sin, position_ids, kp, vp, pt,
                           start_pos)
        hp = (kv.host_prefix(self.layer_index)
              if hasattr(kv, "host_prefix") else None)
        if hp is not None:
            q = (qkv[..., : Hq * D].view(B, T, Hq, D)
                 .permute(0, 2, 1, 3).contiguous())
            attn = ops.attn_paged_mixed(q, kp, vp, pt, start_pos + T,
                                        hp[0], hp[1], self.scale)
            attn = attn.permute(0, 2, 1, 3).reshape(B, T, Hq * D)
        else:
            attn = ops.attn_paged_qkv(qkv, Hq, Hkv, kp, vp, pt, start_pos,
                                      self.scale)
        if fuse_norm:
            self._ensure_ss_bufs(hidden.device)
            h2 = ops.linear(attn, self.o_w, residual=hidden,
                            ss_out=self._ss_h2)
            gu = ops.linear(h2, self.gate_up_w,
                            norm=(None, cfg.rms_norm_eps), ss_in=self._ss_h2)
            out = ops.linear(ops.swiglu(gu), self.down_w, residual=h2,
                             ss_out=self._ss_hidden)
            self._ss_valid = True
            return out
        a = self._lin(attn, self.o_w, "o_w")
        h2, y = ops.rms_norm_residual(a, hidden, self.post_norm_w, cfg.rms_norm_eps)
        return self._lin(ops.swiglu(self._lin(y, self.gate_up_w, "gate_up_w")),
                         self.down_w, "down_w", residual=h2)

    def forward_train(self, hidden: torch.Tensor, start_pos: int = 0) -> torch.Tensor:
        B, T, H = hidden.shape
        Hq, Hkv, D = self.Hq, self.Hkv, self.D
        cfg = self.config
        G = Hq // Hkv

        def rms(x, w):
            xf = x.float()
            return (xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True)
                                     + cfg.rms_norm_eps)).to(x.dtype) * w

        x = rms(hidden, self.input_norm_w)
        qkv = torch.nn.functional.linear(x, self.qkv_w)
        qkv = qkv.view(B, T, Hq + 2 * Hkv, D).permute(0, 2, 1, 3)
        q, k, v = qkv.split([Hq, Hkv, Hkv], dim=1)
        q = rms(q, self.q_norm_w)
        k = rms(k, self.k_norm_w)
        cos, sin = self.rope.get(hidden.device)
        pos = torch.arange(start_pos, start_pos + T).view(1, T).expand(B, T)
        from bloombee_amd.ops import reference as refops
        q, k = refops.rope_apply(q, k, cos, sin, pos)
        k = k.repeat_interleave(G, dim=1)
        v = v.repeat_interleave(G, dim=1)
        scores = torch.matmul(q.float(), k.float().transpose(-1, -2)) * self.scale
        mask = torch.ones(T, T, dtype=torch.bool).tril()
        scores = scores.masked_fill(~mask.to(scores.device), float("-inf"))
        p = torch.softmax(scores, dim=-1)
        attn = torch.matmul(p, v.float()).to(hidden.dtype)
        attn = attn.permute(0, 2, 1, 3).reshape(B, T, Hq * D)
        h2 = hidden + torch.nn.functional.linear(attn, self.o_w)
        y = rms(h2, self.post_norm_w)
        gu = torch.nn.functional.linear(y, self.gate_up_w)
        g, u = gu.split([self.I, self.I], dim=-1)
        m = torch.nn.functional.linear(
            torch.nn.functional.silu(g.float()).to(u.dtype) * u, self.down_w)
        return h2 + m
