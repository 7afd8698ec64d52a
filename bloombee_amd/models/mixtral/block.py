"""Server-side Mixtral block (parity: reference models/mixtral/block.py
WrappedMixtralBlock :13-137): LLaMA-style GQA attention + top-2 MoE MLP.

MoE execution (MI-native): tokens are gathered per selected expert and each
expert's gate_up/down run as skinny-M GEMMs — at decode batch sizes every
expert's token group is M <= 32, exactly the shape gemm_skinny.hip is built
for (BASELINE.json config 4's "MoE grouped GEMM"); the scatter-add back is a
single index_add per expert."""
from __future__ import annotations

import math
from typing import Optional

import torch

from bloombee_amd import ops
from bloombee_amd.kv.paged import SessionHandle
from bloombee_amd.models.base import ModelConfig
from bloombee_amd.models.llama.block import RopeTables


class MixtralBlock(torch.nn.Module):
    def __init__(self, config: ModelConfig, layer_index: int = 0,
                 rope: Optional[RopeTables] = None):
        super().__init__()
        self.config = config
        self.layer_index = layer_index
        H = config.hidden_size
        D = config.head_dim
        Hq, Hkv = config.num_attention_heads, config.num_key_value_heads
        I = config.intermediate_size
        E = int(config.extras.get("num_local_experts", 8))
        K = int(config.extras.get("num_experts_per_tok", 2))
        dt = config.dtype
        self.Hq, self.Hkv, self.D, self.I, self.E, self.topk = Hq, Hkv, D, I, E, K
        self.scale = 1.0 / math.sqrt(D)
        self.rope = rope if rope is not None else RopeTables(config)

        def p(*shape):
            return torch.nn.Parameter(torch.empty(*shape, dtype=dt),
                                      requires_grad=False)

        self.input_norm_w = p(H)
        self.qkv_w = p((Hq + 2 * Hkv) * D, H)
        self.o_w = p(H, Hq * D)
        self.post_norm_w = p(H)
        self.router_w = p(E, H)
        self.expert_gate_up_w = p(E, 2 * I, H)
        self.expert_down_w = p(E, H, I)

    @torch.no_grad()
    def init_random(self, seed: Optional[int] = None):
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

    # -- MoE MLP ----------------------------------------------------------
    def _moe(self, y: torch.Tensor) -> torch.Tensor:
        B, T, H = y.shape
        flat = y.reshape(-1, H)
        logits = ops.linear(flat, self.router_w).float()
        weights, experts = logits.topk(self.topk, dim=-1)
        weights = torch.softmax(weights, dim=-1).to(y.dtype)
        e0, e1 = getattr(self, "ep_range", (0, self.E))
        if (flat.is_cuda and ops.HAVE_HIP_OPS
                and flat.dtype == torch.bfloat16
                and (e0, e1) == (0, self.E) and flat.shape[0] <= 1024
                and not getattr(self, "moe_loop", False)):
            # grouped decode path: expert-sort the (token, choice) slots on
            # device and run BOTH expert GEMMs as single grouped launches
            # (ops/hip/moe_gemm.hip) — no per-expert host sync, no
            # per-expert kernel launches (BASELINE config 4)
            Tk = flat.shape[0] * self.topk
            fe = experts.reshape(-1)
            order = fe.argsort(stable=True)
            counts = torch.bincount(fe, minlength=self.E)
            off = torch.zeros(self.E + 1, dtype=torch.int32,
                              device=flat.device)
            off[1:] = counts.cumsum(0).int()
            tok = (order // self.topk).int()
            wsorted = weights.reshape(-1)[order].float()
            mch = (flat.shape[0] + 31) // 32
            gu = ops.moe_gemm_grouped(flat, self.expert_gate_up_w, off,
                                      rowmap=tok, S=Tk, mchunks=mch)
            dn = ops.moe_gemm_grouped(ops.swiglu(gu), self.expert_down_w,
                                      off, scale=wsorted, S=Tk, mchunks=mch)
            out = torch.zeros_like(flat)
            out.index_add_(0, tok.long(), dn)
            return out.view(B, T, H)
        out = torch.zeros_like(flat)
        # expert-parallel mode (parallel/expert.py): this rank holds only
        # experts [e0, e1); partial sums cross ranks via one all-reduce
        for le, e in enumerate(range(e0, e1)):
            sel = (experts == e)
            rows = sel.any(dim=-1).nonzero(as_tuple=True)[0]
            if rows.numel() == 0:
                continue
            xe = flat[rows]
            h = ops.linear(ops.swiglu(ops.linear(xe, self.expert_gate_up_w[le])),
                           self.expert_down_w[le])
            w = (weights * sel.to(weights.dtype)).sum(-1)[rows]
            out.index_add_(0, rows, h * w.unsqueeze(-1))
        if getattr(self, "ep_enabled", False):
            from bloombee_amd.parallel.expert import ep_all_reduce
            ep_all_reduce(out, self.ep_group)
        return out.view(B, T, H)

    @torch.no_grad()
    def forward_inference(self, hidden: torch.Tensor, kv: SessionHandle,
                          start_pos: torch.Tensor,
                          position_ids=None) -> torch.Tensor:
        B, T, H = hidden.shape
        Hq, Hkv = self.Hq, self.Hkv
        cfg = self.config
        x = ops.rms_norm(hidden, self.input_norm_w, cfg.rms_norm_eps)
        qkv = ops.linear(x, self.qkv_w)
        cos, sin = self.rope.get(hidden.device)
        kp = kv.k_pages(self.layer_index)
        vp = kv.v_pages(self.layer_index)
        pt = kv.page_table()
        ops.rope_kv_write_(qkv, Hq, Hkv, cos, sin, position_ids, kp, vp, pt,
                           start_pos)
        attn = ops.attn_paged_qkv(qkv, Hq, Hkv, kp, vp, pt, start_pos, self.scale)
        a = ops.linear(attn, self.o_w)
        h2, y = ops.rms_norm_residual(a, hidden, self.post_norm_w, cfg.rms_norm_eps)
        return h2 + self._moe(y)

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

        flat = y.reshape(-1, H)
        logits = torch.nn.functional.linear(flat, self.router_w).float()
        weights, experts = logits.topk(self.topk, dim=-1)
        weights = torch.softmax(weights, dim=-1).to(y.dtype)
        out = torch.zeros_like(flat)
        e0, e1 = getattr(self, "ep_range", (0, self.E))
        for le, e in enumerate(range(e0, e1)):
            sel = (experts == e)
            rows = sel.any(dim=-1).nonzero(as_tuple=True)[0]
            if rows.numel() == 0:
                continue
            xe = flat[rows]
            gu = torch.nn.functional.linear(xe, self.expert_gate_up_w[le])
            g, u = gu.split([self.I, self.I], dim=-1)
            h = torch.nn.functional.linear(
                torch.nn.functional.silu(g.float()).to(u.dtype) * u,
                self.expert_down_w[le])
            w = (weights * sel.to(weights.dtype)).sum(-1)[rows]
            out = out.index_add(0, rows, h * w.unsqueeze(-1))
        if getattr(self, "ep_enabled", False):
            from bloombee_amd.parallel.expert import ep_all_reduce
            ep_all_reduce(out, self.ep_group)
        return h2 + out.view(B, T, H)

    def forward(self, *args, **kw):
        return self.forward_inference(*args, **kw)
