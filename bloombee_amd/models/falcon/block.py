"""Server-side Falcon block (parity: reference models/falcon/block.py
WrappedFalconBlock :399-503 — falcon-7b architecture: single input LayerNorm,
parallel attention + MLP branches, rotary MQA (num_kv_heads=1), no biases).

MQA note: GQA group size Hq/Hkv = 71 exceeds the decode kernel's 16-row MFMA
q-tile, so the GPU path reshapes queries into ceil(G/16) fake groups sharing
the one KV head at the interface level (CPU reference handles any G)."""
from __future__ import annotations

import math
from typing import Optional

import torch

from bloombee_amd import ops
from bloombee_amd.kv.paged import SessionHandle
from bloombee_amd.models.base import ModelConfig
from bloombee_amd.models.llama.block import RopeTables


class FalconBlock(torch.nn.Module):
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
        self.rope = rope if rope is not None else RopeTables(config)

        def p(*shape):
            return torch.nn.Parameter(torch.empty(*shape, dtype=dt),
                                      requires_grad=False)

        self.ln_w, self.ln_b = p(H), p(H)
        self.qkv_w = p((Hq + 2 * Hkv) * D, H)
        self.o_w = p(H, Hq * D)
        self.up_w = p(I, H)
        self.down_w = p(H, I)

    @torch.no_grad()
    def init_random(self, seed: Optional[int] = None):
        s = seed if seed is not None else 1234 + self.layer_index
        dev = self.ln_w.device
        gen = torch.Generator(device=dev).manual_seed(s)
        std = 0.02 / math.sqrt(2 * self.config.num_hidden_layers)
        for name, w in self.named_parameters():
            if name == "ln_b":
                w.zero_()
            elif name == "ln_w":
                w.fill_(1.0)
            else:
                w.copy_(torch.randn(w.shape, generator=gen, dtype=torch.float32,
                                    device=dev).mul_(std).to(w.dtype))
        return self

    @torch.no_grad()
    def forward_inference(self, hidden: torch.Tensor, kv: SessionHandle,
                          start_pos: torch.Tensor,
                          position_ids=None) -> torch.Tensor:
        B, T, H = hidden.shape
        Hq, Hkv, D = self.Hq, self.Hkv, self.D
        cfg = self.config
        x = ops.layer_norm(hidden, self.ln_w, self.ln_b, cfg.layer_norm_epsilon)
        qkv = ops.linear(x, self.qkv_w)
        cos, sin = self.rope.get(hidden.device)
        kp = kv.k_pages(self.layer_index)
        vp = kv.v_pages(self.layer_index)
        pt = kv.page_table()
        ops.rope_kv_write_(qkv, Hq, Hkv, cos, sin, position_ids, kp, vp, pt,
                           start_pos)
        attn = ops.attn_paged_qkv(qkv, Hq, Hkv, kp, vp, pt, start_pos,
                                  self.scale)
        a = ops.linear(attn, self.o_w)
        # parallel branches: both read the SAME layernorm output
        m = ops.linear(ops.gelu_tanh(ops.linear(x, self.up_w)), self.down_w)
        return hidden + a + m

    def forward_train(self, hidden: torch.Tensor, start_pos: int = 0) -> torch.Tensor:
        B, T, H = hidden.shape
        Hq, Hkv, D = self.Hq, self.Hkv, self.D
        cfg = self.config
        G = Hq // Hkv
        x = torch.nn.functional.layer_norm(
            hidden.float(), (H,), self.ln_w.float(), self.ln_b.float(),
            cfg.layer_norm_epsilon).to(hidden.dtype)
        qkv = torch.nn.functional.linear(x, self.qkv_w)
        q = qkv[..., :Hq * D].view(B, T, Hq, D).permute(0, 2, 1, 3)
        k = qkv[..., Hq * D:(Hq + Hkv) * D].view(B, T, Hkv, D).permute(0, 2, 1, 3)
        v = qkv[..., (Hq + Hkv) * D:].view(B, T, Hkv, D).permute(0, 2, 1, 3)
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
        a = torch.nn.functional.linear(attn, self.o_w)
        u = torch.nn.functional.linear(x, self.up_w)
        g = torch.nn.functional.gelu(u.float(), approximate="tanh").to(u.dtype)
        m = torch.nn.functional.linear(g, self.down_w)
        return hidden + a + m

    def forward(self, *args, **kw):
        return self.forward_inference(*args, **kw)
