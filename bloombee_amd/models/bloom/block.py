"""Server-side BLOOM block (parity: reference models/bloom/block.py
WrappedBloomBlock — HF eager with alibi, :108-188). MI-native composition on
the shared op layer: LayerNorm + fused-QKV + alibi paged attention +
gelu MLP, all biased; MHA (num_key_value_heads == num_attention_heads);
KV rides the same paged pools as every family."""
from __future__ import annotations

import math
from typing import Optional

import torch

from bloombee_amd import ops
from bloombee_amd.kv.paged import SessionHandle
from bloombee_amd.models.base import ModelConfig


class BloomBlock(torch.nn.Module):
    def __init__(self, config: ModelConfig, layer_index: int = 0, rope=None):
        super().__init__()
        self.config = config
        self.layer_index = layer_index
        H = config.hidden_size
        D = config.head_dim
        Hq = config.num_attention_heads
        I = config.intermediate_size
        dt = config.dtype
        self.Hq, self.D, self.I = Hq, D, I
        self.scale = 1.0 / math.sqrt(D)
        # buffer (not a plain attr) so .to(device) moves it once: a CPU
        # tensor here means a pageable H2D copy per layer per step in the
        # attention dispatch — wasteful eager and illegal under hipGraph
        # capture (the bloom capture fault isolated in r02)
        self.register_buffer("alibi", ops.alibi_slopes_for(Hq).float(),
                             persistent=False)

        def p(*shape):
            return torch.nn.Parameter(torch.empty(*shape, dtype=dt),
                                      requires_grad=False)

        self.ln1_w, self.ln1_b = p(H), p(H)
        self.qkv_w, self.qkv_b = p(3 * Hq * D, H), p(3 * Hq * D)
        self.dense_w, self.dense_b = p(H, Hq * D), p(H)
        self.ln2_w, self.ln2_b = p(H), p(H)
        self.up_w, self.up_b = p(I, H), p(I)
        self.down_w, self.down_b = p(H, I), p(H)

    @torch.no_grad()
    def init_random(self, seed: Optional[int] = None):
        s = seed if seed is not None else 1234 + self.layer_index
        dev = self.ln1_w.device
        gen = torch.Generator(device=dev).manual_seed(s)
        std = 0.02 / math.sqrt(2 * self.config.num_hidden_layers)
        for name, w in self.named_parameters():
            if name.endswith("_b"):
                w.zero_()
            elif name.startswith("ln"):
                w.fill_(1.0)
            else:
                w.copy_(torch.randn(w.shape, generator=gen, dtype=torch.float32,
                                    device=dev).mul_(std).to(w.dtype))
        return self

    def _attn(self, x: torch.Tensor, kv: SessionHandle,
              start_pos: torch.Tensor) -> torch.Tensor:
        B, T, _ = x.shape
        Hq, D = self.Hq, self.D
        qkv = ops.linear(x, self.qkv_w, bias=self.qkv_b)
        q = qkv[..., :Hq * D].view(B, T, Hq, D).permute(0, 2, 1, 3).contiguous()
        k = qkv[..., Hq * D:2 * Hq * D].view(B, T, Hq, D).permute(0, 2, 1, 3).contiguous()
        v = qkv[..., 2 * Hq * D:].view(B, T, Hq, D).permute(0, 2, 1, 3).contiguous()
        kp = kv.k_pages(self.layer_index)
        vp = kv.v_pages(self.layer_index)
        pt = kv.page_table()
        ops.kv_write(k, v, kp, vp, pt, start_pos)
        out = ops.attn_paged(q, kp, vp, pt, start_pos.long(), self.scale,
                             alibi_slopes=self.alibi)
        out = out.permute(0, 2, 1, 3).reshape(B, T, Hq * D)
        return ops.linear(out, self.dense_w, bias=self.dense_b)

    @torch.no_grad()
    def forward_inference(self, hidden: torch.Tensor, kv: SessionHandle,
                          start_pos: torch.Tensor,
                          position_ids=None) -> torch.Tensor:
        cfg = self.config
        x = ops.layer_norm(hidden, self.ln1_w, self.ln1_b, cfg.layer_norm_epsilon)
        h = hidden + self._attn(x, kv, start_pos)
        y = ops.layer_norm(h, self.ln2_w, self.ln2_b, cfg.layer_norm_epsilon)
        m = ops.linear(ops.gelu_tanh(ops.linear(y, self.up_w, bias=self.up_b)),
                       self.down_w, bias=self.down_b)
        return h + m

    def forward_train(self, hidden: torch.Tensor, start_pos: int = 0) -> torch.Tensor:
        cfg = self.config
        B, T, H = hidden.shape
        Hq, D = self.Hq, self.D

        def ln(x, w, b):
            return torch.nn.functional.layer_norm(
                x.float(), (H,), w.float(), b.float(),
                cfg.layer_norm_epsilon).to(x.dtype)

        x = ln(hidden, self.ln1_w, self.ln1_b)
        qkv = torch.nn.functional.linear(x, self.qkv_w, self.qkv_b)
        q, k, v = (t.view(B, T, Hq, D).permute(0, 2, 1, 3)
                   for t in qkv.split(Hq * D, dim=-1))
        scores = torch.matmul(q.float(), k.float().transpose(-1, -2)) * self.scale
        pos = torch.arange(start_pos, start_pos + T)
        scores = scores + self.alibi.view(1, Hq, 1, 1) * \
            (start_pos + torch.arange(T, dtype=torch.float32)).view(1, 1, 1, T)
        mask = torch.ones(T, T, dtype=torch.bool).tril()
        scores = scores.masked_fill(~mask.to(scores.device), float("-inf"))
        p = torch.softmax(scores, dim=-1)
        attn = torch.matmul(p, v.float()).to(hidden.dtype)
        attn = attn.permute(0, 2, 1, 3).reshape(B, T, Hq * D)
        h = hidden + torch.nn.functional.linear(attn, self.dense_w, self.dense_b)
        y = ln(h, self.ln2_w, self.ln2_b)
        u = torch.nn.functional.linear(y, self.up_w, self.up_b)
        g = torch.nn.functional.gelu(u.float(), approximate="tanh").to(u.dtype)
        m = torch.nn.functional.linear(g, self.down_w, self.down_b)
        return h + m

    def forward(self, *args, **kw):
        return self.forward_inference(*args, **kw)
