"""Server-side Gemma-4 block (parity: reference models/gemma4/block.py
WrappedGemma4Block :38-261). See config.py for the family's heterogeneity;
every norm is plain-weight RMS (Gemma-4 dropped the (1+w) convention);
attention scaling is 1.0."""
from __future__ import annotations

import math
from typing import Optional

import torch

from bloombee_amd import ops
from bloombee_amd.kv.paged import SessionHandle
from bloombee_amd.models.gemma4.config import Gemma4Config
from bloombee_amd.ops.reference import rope_cos_sin


def _rms1p(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    # Gemma-4 dropped the older gemma (1+weight) RMS convention: plain weight,
    # ones-initialized (HF Gemma4RMSNorm).
    return ops.rms_norm(x, w, eps)


def _rms1p_headdim(x: torch.Tensor, w: Optional[torch.Tensor], eps: float):
    """Per-head RMS over the last dim; w None => scale-less (v_norm)."""
    xf = x.float()
    y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    if w is not None:
        y = y * w.float()
    return y.to(x.dtype)


def _rope_partial(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                  pos: torch.Tensor, rot_dim: int) -> torch.Tensor:
    """Rotate the first rot_dim dims of x (B, H, T, D); rest pass through."""
    xf = x.float()
    xr, xp = xf[..., :rot_dim], xf[..., rot_dim:]
    c = cos[pos.long()].unsqueeze(1).float()
    s = sin[pos.long()].unsqueeze(1).float()
    x1, x2 = xr[..., : rot_dim // 2], xr[..., rot_dim // 2:]
    out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s, xp], dim=-1)
    return out.to(x.dtype)


class Gemma4Block(torch.nn.Module):
    def __init__(self, config: Gemma4Config, layer_index: int = 0, rope=None):
        super().__init__()
        self.config = config
        self.layer_index = layer_index          # global at init; BlockStack
        self.global_index = layer_index         # rebases layer_index locally
        H = config.hidden_size
        self.Hq = config.num_attention_heads
        self.Hkv = config.hkv_for(layer_index)
        self.D = config.head_dim_for(layer_index)
        self.window = config.window_for(layer_index)
        theta, prf = config.rope_for(layer_index)
        self.rot_dim = int(prf * self.D) // 2 * 2
        self.I = config.intermediate_size
        self.k_eq_v = bool(config.extras.get("attention_k_eq_v", False))
        self.donor = config.shared_kv_donor(layer_index)
        dt = config.dtype
        Hq, Hkv, D = self.Hq, self.Hkv, self.D

        cos, sin = rope_cos_sin(self.rot_dim, config.max_position_embeddings,
                                theta=theta)
        self._cos, self._sin = cos, sin
        self._rope_cache = {}

        def p(*shape):
            return torch.nn.Parameter(torch.empty(*shape, dtype=dt),
                                      requires_grad=False)

        self.input_norm_w = p(H)
        self.post_attn_norm_w = p(H)
        self.pre_ffn_norm_w = p(H)
        self.post_ffn_norm_w = p(H)
        self.q_w = p(Hq * D, H)
        self.q_norm_w = p(D)
        if self.donor is None:
            self.k_w = p(Hkv * D, H)
            self.k_norm_w = p(D)
            if not (self.k_eq_v and self.window == 0):
                self.v_w = p(Hkv * D, H)
            else:
                self.v_w = None
        else:
            self.k_w = self.v_w = None
        self.o_w = p(H, Hq * D)
        self.gate_up_w = p(2 * self.I, H)
        self.down_w = p(H, self.I)
        if int(config.extras.get("hidden_size_per_layer_input", 0)):
            raise NotImplementedError(
                "gemma-4 per-layer-input path not implemented")

    def _tables(self, device):
        key = str(device)
        if key not in self._rope_cache:
            self._rope_cache[key] = (self._cos.to(device), self._sin.to(device))
        return self._rope_cache[key]

    @torch.no_grad()
    def init_random(self, seed: Optional[int] = None):
        s = seed if seed is not None else 1234 + self.global_index
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

    @torch.no_grad()
    def forward_inference(self, hidden: torch.Tensor, kv: SessionHandle,
                          start_pos: torch.Tensor,
                          position_ids=None) -> torch.Tensor:
        B, T, H = hidden.shape
        Hq, Hkv, D = self.Hq, self.Hkv, self.D
        cfg = self.config
        eps = cfg.rms_norm_eps
        cos, sin = self._tables(hidden.device)
        if position_ids is None:
            pos = start_pos.view(B, 1).long() + torch.arange(
                T, device=hidden.device).view(1, T)
        else:
            pos = position_ids.long()

        x = _rms1p(hidden, self.input_norm_w, eps)
        q = ops.linear(x, self.q_w).view(B, T, Hq, D).permute(0, 2, 1, 3)
        q = _rms1p_headdim(q, self.q_norm_w, eps)
        q = _rope_partial(q, cos, sin, pos, self.rot_dim).contiguous()

        if self.donor is None:
            kraw = ops.linear(x, self.k_w).view(B, T, Hkv, D).permute(0, 2, 1, 3)
            k = _rms1p_headdim(kraw, self.k_norm_w, eps)
            k = _rope_partial(k, cos, sin, pos, self.rot_dim)
            if self.v_w is not None:
                vraw = ops.linear(x, self.v_w).view(B, T, Hkv, D).permute(0, 2, 1, 3)
            else:
                vraw = kraw                      # attention_k_eq_v
            v = _rms1p_headdim(vraw, None, eps)  # scale-less v_norm
            kv_layer = self.layer_index
            ops.kv_write(k.contiguous(), v.contiguous(),
                         kv.k_pages(kv_layer), kv.v_pages(kv_layer),
                         kv.page_table(), start_pos)
        else:
            # shared-KV tail: read the donor layer's pools (donor must be
            # co-hosted; BlockStack rebased layer_index by the same offset)
            kv_layer = self.layer_index - (self.global_index - self.donor)
            if kv_layer < 0:
                raise RuntimeError(
                    f"gemma-4 shared-KV layer {self.global_index} requires its "
                    f"donor layer {self.donor} in the same server block range")

        attn = ops.attn_paged(q, kv.k_pages(kv_layer), kv.v_pages(kv_layer),
                              kv.page_table(), start_pos.long(), scale=1.0,
                              window=self.window)
        attn = attn.permute(0, 2, 1, 3).reshape(B, T, Hq * D)
        a = ops.linear(attn, self.o_w)
        a = _rms1p(a, self.post_attn_norm_w, eps)
        h = hidden + a

        y = _rms1p(h, self.pre_ffn_norm_w, eps)
        gu = ops.linear(y, self.gate_up_w)
        g, u = gu.split([self.I, self.I], dim=-1)
        m = ops.linear(ops.gelu_tanh(g) * u, self.down_w)
        m = _rms1p(m, self.post_ffn_norm_w, eps)
        return h + m

    def forward_train(self, hidden: torch.Tensor, start_pos: int = 0) -> torch.Tensor:
        """Differentiable full-sequence path (no KV). Shared-KV layers need
        the donor's k/v, recomputed here is impossible without its weights —
        forward_train therefore requires donor-free configs (the fine-tuning
        RPCs run span-wise where the constraint holds by construction)."""
        if self.donor is not None:
            raise NotImplementedError(
                "training path through gemma-4 shared-KV layers")
        B, T, H = hidden.shape
        Hq, Hkv, D = self.Hq, self.Hkv, self.D
        eps = self.config.rms_norm_eps
        cos, sin = self._tables(hidden.device)
        pos = torch.arange(start_pos, start_pos + T,
                           device=hidden.device).view(1, T).expand(B, T)
        G = Hq // Hkv

        x = _rms1p(hidden, self.input_norm_w, eps)
        q = torch.nn.functional.linear(x, self.q_w).view(B, T, Hq, D).permute(0, 2, 1, 3)
        q = _rms1p_headdim(q, self.q_norm_w, eps)
        q = _rope_partial(q, cos, sin, pos, self.rot_dim)
        kraw = torch.nn.functional.linear(x, self.k_w).view(B, T, Hkv, D).permute(0, 2, 1, 3)
        k = _rms1p_headdim(kraw, self.k_norm_w, eps)
        k = _rope_partial(k, cos, sin, pos, self.rot_dim)
        vraw = (torch.nn.functional.linear(x, self.v_w)
                .view(B, T, Hkv, D).permute(0, 2, 1, 3)
                if self.v_w is not None else kraw)
        v = _rms1p_headdim(vraw, None, eps)
        k = k.repeat_interleave(G, dim=1)
        v = v.repeat_interleave(G, dim=1)
        scores = torch.matmul(q.float(), k.float().transpose(-1, -2))  # scale 1.0
        qpos = torch.arange(start_pos, start_pos + T).view(T, 1)
        kpos = torch.arange(start_pos, start_pos + T).view(1, T)
        mask = kpos <= qpos
        if self.window > 0:
            mask &= kpos > qpos - self.window
        scores = scores.masked_fill(~mask.to(scores.device), float("-inf"))
        p = torch.softmax(scores, dim=-1)
        attn = torch.matmul(p, v.float()).to(hidden.dtype)
        attn = attn.permute(0, 2, 1, 3).reshape(B, T, Hq * D)
        a = torch.nn.functional.linear(attn, self.o_w)
        a = _rms1p(a, self.post_attn_norm_w, eps)
        h = hidden + a
        y = _rms1p(h, self.pre_ffn_norm_w, eps)
        gu = torch.nn.functional.linear(y, self.gate_up_w)
        g, u = gu.split([self.I, self.I], dim=-1)
        gact = torch.nn.functional.gelu(g.float(), approximate="tanh").to(u.dtype)
        m = torch.nn.functional.linear(gact * u, self.down_w)
        m = _rms1p(m, self.post_ffn_norm_w, eps)
        return h + m

    def forward(self, *args, **kw):
        return self.forward_inference(*args, **kw)
