"""Local (single-process) inference engine: embeddings + block stack + LM head.

This is the innermost compute loop shared by:
  * bench.py at N=1 (the whole model on one MI355X),
  * the per-worker stage runner in parallel/pipeline.py (a contiguous block
    range per rank),
  * the server backend (server/backend.py) which serves the same blocks over
    the decentralized transport.

The reference splits this across DistributedLlamaModel (client embeddings +
lm_head, models/llama/model.py:45-118) and per-server block execution; here
the same Blocks object serves both local and distributed paths.
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import torch
import torch.nn.functional as F

from bloombee_amd.kv.paged import PagedKVCache, SessionHandle
from bloombee_amd.models.auto import get_block_class
from bloombee_amd.models.base import ModelConfig, resolve_config
from bloombee_amd.models.llama.block import RopeTables
from bloombee_amd.utils.logging import get_logger
from bloombee_amd import ops

logger = get_logger(__name__)


class BlockStack(torch.nn.Module):
    """A contiguous range [start, end) of transformer blocks on one device."""

    def __init__(self, config: ModelConfig, start: int, end: int,
                 device="cpu", seed: int = 0):
        super().__init__()
        self.config = config
        self.start, self.end = start, end
        block_cls = get_block_class(config.model_type)
        self.rope = RopeTables(config)
        self.blocks = torch.nn.ModuleList()
        for i in range(start, end):
            blk = block_cls(config, layer_index=i, rope=self.rope).to(device)
            blk.init_random(seed=seed * 10_000 + i)
            blk.layer_index = i - start  # KV page index local to this stack's pool
            self.blocks.append(blk)
        # chain blocks so each one's down-proj epilogue can leave row
        # sum-of-squares for the next one's fused input rmsnorm
        # (llama/block.py fuse_norm path); first block has no producer
        # and keeps the separate-norm launch
        for j in range(1, len(self.blocks)):
            # bypass Module.__setattr__: a plain reference, not a submodule
            object.__setattr__(self.blocks[j], "prev_block",
                               self.blocks[j - 1])
        self.device = torch.device(device)

    def make_kv(self, max_tokens: int) -> PagedKVCache:
        # per-layer KV geometry hook (gemma-4 heterogeneous head_dim/kv heads)
        if hasattr(self.config, "kv_geometry"):
            hkv, dims = self.config.kv_geometry(self.start, self.end)
        else:
            hkv = self.config.num_key_value_heads
            dims = self.config.head_dim
        from bloombee_amd.config import get_config
        P = get_config().kv.page_size
        # the decode kernels walk KV in 32-position tiles and require page
        # boundaries to align (attn_decode_mfma.hip: "P | 32")
        if 32 % P != 0:
            raise ValueError(f"BBAMD_KV_PAGE_SIZE must divide 32, got {P}")
        return PagedKVCache(
            num_layers=len(self.blocks),
            num_kv_heads=hkv,
            head_dim=dims,
            max_tokens=max_tokens,
            page_size=P,
            device=self.device,
            dtype=self.config.dtype,
        )

    @torch.no_grad()
    def forward_inference(self, hidden: torch.Tensor, kv: SessionHandle,
                          start_pos: torch.Tensor,
                          position_ids: Optional[torch.Tensor] = None,
                          tree_mask: Optional[torch.Tensor] = None,
                          deep_prompts: Optional[torch.Tensor] = None,
                          ) -> torch.Tensor:
        # deep p-tuning (ref client/ptune.py deep mode): deep_prompts is
        # (n_local_blocks, pre, H); block i's slice is added to the prompt
        # positions (< pre) of its input. Decode steps past the prompt
        # region skip the addition entirely.
        sp0 = int(start_pos[0]) if deep_prompts is not None else 0
        pre = deep_prompts.shape[1] if deep_prompts is not None else 0
        n_over = min(pre - sp0, hidden.shape[1]) if sp0 < pre else 0
        for i, blk in enumerate(self.blocks):
            if n_over > 0:
                dp = deep_prompts[i, sp0:sp0 + n_over].to(hidden.dtype)
                hidden[:, :n_over] += dp
            if tree_mask is not None:
                hidden = blk.forward_inference(hidden, kv, start_pos,
                                               position_ids, tree_mask=tree_mask)
            else:
                hidden = blk.forward_inference(hidden, kv, start_pos, position_ids)
        return hidden

    def forward_train(self, hidden: torch.Tensor, start_pos: int = 0,
                      deep_prompts: Optional[torch.Tensor] = None) -> torch.Tensor:
        pre = deep_prompts.shape[1] if deep_prompts is not None else 0
        for i, blk in enumerate(self.blocks):
            if pre > 0 and hidden.shape[1] >= pre:
                # functional (not in-place) so prompt grads flow in backward
                dp = deep_prompts[i].to(hidden.dtype).unsqueeze(0)
                hidden = torch.cat([hidden[:, :pre] + dp, hidden[:, pre:]], 1)
            hidden = blk.forward_train(hidden, start_pos)
        return hidden


class LocalEngine:
    """Whole model in one process: greedy decode for bench/tests.

    Client-side pieces (embeddings, final norm, LM head) follow the reference
    client split (client/lm_head.py, models/llama/model.py:80-118).
    """

    def __init__(self, config_or_name, device="cpu", seed: int = 0,
                 kv_max_tokens: int = 1 << 16, quantize_q4: bool = False):
        cfg = (config_or_name if isinstance(config_or_name, ModelConfig)
               else resolve_config(config_or_name))
        self.config = cfg
        self.device = torch.device(device)
        gen = torch.Generator().manual_seed(seed)
        dt = cfg.dtype
        # family head shape: bloom/falcon use LayerNorm heads; bloom also
        # layer-norms the embedding output (HF word_embeddings_layernorm)
        self.embed_ln = cfg.model_type == "bloom"
        self.ln_final = cfg.model_type in ("bloom", "falcon")
        self.embed_scale = (cfg.hidden_size ** 0.5
                            if cfg.model_type == "gemma4" else 1.0)
        self.embed = (torch.randn(cfg.vocab_size, cfg.hidden_size, generator=gen)
                      .mul_(0.02).to(dt).to(device))
        self.final_norm_w = torch.ones(cfg.hidden_size, dtype=dt, device=device)
        self.final_norm_b = torch.zeros(cfg.hidden_size, dtype=dt, device=device)
        if self.embed_ln:
            self.embed_ln_w = torch.ones(cfg.hidden_size, dtype=dt, device=device)
            self.embed_ln_b = torch.zeros(cfg.hidden_size, dtype=dt, device=device)
        if cfg.tie_word_embeddings:
            self.lm_head_w = self.embed
        else:
            self.lm_head_w = (torch.randn(cfg.vocab_size, cfg.hidden_size, generator=gen)
                              .mul_(0.02).to(dt).to(device))
        self.stack = BlockStack(cfg, 0, cfg.num_hidden_layers, device=device, seed=seed)
        if quantize_q4:
            # 4-bit weight-compressed decode (w4_gemm.hip): ~4x less weight
            # memory + HBM traffic. Used for cheap same-weights draft models
            # (speculative decoding) and compressed serving.
            for blk in self.stack.blocks:
                if hasattr(blk, "quantize_weights_q4"):
                    blk.quantize_weights_q4()
        self.kv_pool = self.stack.make_kv(kv_max_tokens)

    @torch.no_grad()
    def logits_for(self, hidden_last: torch.Tensor) -> torch.Tensor:
        """hidden_last: (B, H) -> (B, V)"""
        if self.ln_final:
            y = ops.layer_norm(hidden_last, self.final_norm_w, self.final_norm_b,
                               self.config.layer_norm_epsilon)
        else:
            y = ops.rms_norm(hidden_last, self.final_norm_w, self.config.rms_norm_eps)
        return F.linear(y, self.lm_head_w)

    def _embed(self, input_ids: torch.Tensor) -> torch.Tensor:
        h = F.embedding(input_ids.to(self.device), self.embed)
        if self.embed_scale != 1.0:
            h = h * self.embed_scale
        if self.embed_ln:
            h = ops.layer_norm(h, self.embed_ln_w, self.embed_ln_b,
                               self.config.layer_norm_epsilon)
        return h

    @torch.no_grad()
    def prefill(self, input_ids: torch.Tensor, kv: SessionHandle) -> torch.Tensor:
        """input_ids: (B, T). Returns greedy next token ids (B,)."""
        B, T = input_ids.shape
        start = torch.tensor([s.l_spec for s in kv.seqs], dtype=torch.int32,
                             device=self.device)
        kv.extend(T)
        hidden = self._embed(input_ids)
        hidden = self.stack.forward_inference(hidden, kv, start)
        return self.logits_for(hidden[:, -1]).argmax(-1)

    @torch.no_grad()
    def decode_step(self, input_ids: torch.Tensor, kv: SessionHandle) -> torch.Tensor:
        """input_ids: (B,) last tokens. Returns next token ids (B,)."""
        start = torch.tensor([s.l_spec for s in kv.seqs], dtype=torch.int32,
                             device=self.device)
        kv.extend(1)
        hidden = self._embed(input_ids.view(-1, 1))
        hidden = self.stack.forward_inference(hidden, kv, start)
        return self.logits_for(hidden[:, -1]).argmax(-1)

    @torch.no_grad()
    def generate_greedy(self, input_ids: torch.Tensor, max_new_tokens: int) -> torch.Tensor:
        B, T = input_ids.shape
        kv = self.kv_pool.allocate(B, T + max_new_tokens + 1)
        try:
            toks = [self.prefill(input_ids, kv)]
            for _ in range(max_new_tokens - 1):
                toks.append(self.decode_step(toks[-1], kv))
            return torch.stack(toks, dim=1)
        finally:
            kv.close()

    def make_graphed_decoder(self, kv: SessionHandle):
        """Single-token GREEDY decode step (B=1) as a hipGraph replay
        against a FIXED session handle. The whole step — embed, blocks,
        logits, softmax, (prob, argmax) — is inside the graph; the host
        writes (token, position) into one pinned staging buffer, replays,
        and reads ONE 8-byte pair back. Measured motivation
        (benchmarks/gstep_micro.py): the eager step costs ~350 µs/node
        and even post-replay eager softmax/argmax ops add ~100 µs EACH —
        fatal for small draft models in speculative decoding.

        Returns step(token:int, position:int) -> (next_tok:int,
        prob:float, probs_buffer). probs_buffer is the graph-owned softmax
        output — clone it before the next replay if kept. The caller owns
        kv growth (extend/rollback) BEFORE each step. Falls back to an
        eager closure off-GPU."""
        from bloombee_amd.config import get_config

        on_gpu = (self.device.type == "cuda"
                  and get_config().use_hip_graphs)
        staging = torch.zeros(2, dtype=torch.long,
                              pin_memory=on_gpu)
        dev_in = torch.zeros(2, dtype=torch.long, device=self.device)

        def _body():
            ids = dev_in[0:1].view(1, 1)
            pos = dev_in[1:2].to(torch.int32)
            h = self._embed(ids)
            h = self.stack.forward_inference(h, kv, pos)
            logits = self.logits_for(h[:, -1]).float()
            p = torch.softmax(logits[0], -1)
            pv, ti = p.max(-1)
            return p, torch.stack((ti.to(torch.float32), pv))

        if not on_gpu:
            def step_eager(tok: int, position: int):
                dev_in[0] = tok
                dev_in[1] = position
                p, pair = _body()
                return int(pair[0]), float(pair[1]), p
            return step_eager

        # warm the allocator/kernels, then capture (thread_local keeps other
        # threads' work — e.g. the channels pump — legal during capture)
        for _ in range(2):
            kv.extend(1, speculative=True)
            kv.page_table()
            _body()
            kv.rollback()
        graph = torch.cuda.CUDAGraph()
        kv.extend(1, speculative=True)
        kv.page_table()
        with torch.cuda.graph(graph, capture_error_mode="thread_local"):
            p_buf, pair_buf = _body()
        kv.rollback()
        pair_host = torch.zeros(2, dtype=torch.float32, pin_memory=True)

        def step(tok: int, position: int):
            staging[0] = tok
            staging[1] = position
            dev_in.copy_(staging, non_blocking=True)
            kv.page_table()  # flush any new page ids
            graph.replay()
            pair_host.copy_(pair_buf, non_blocking=False)  # syncs
            return int(pair_host[0]), float(pair_host[1]), p_buf

        return step

    def make_graphed_chain(self, kv: SessionHandle, depth: int):
        """A whole greedy draft CHAIN (depth single-token steps, each step's
        argmax fed on-device into the next embed) as ONE hipGraph replay —
        one host round-trip per drafting round instead of per node (the
        per-node graphed step still costs ~150 µs of host floor,
        benchmarks/gstep_micro.py).

        Caller per round: rollback, extend(depth, speculative=True),
        page_table(), then chain(token, position0) -> (tokens[depth],
        probs[depth]) on the host. Off-GPU returns None (callers fall back
        to per-node stepping)."""
        from bloombee_amd.config import get_config

        if self.device.type != "cuda" or not get_config().use_hip_graphs:
            return None
        staging = torch.zeros(2, dtype=torch.long, pin_memory=True)
        dev_in = torch.zeros(2, dtype=torch.long, device=self.device)
        out_host = torch.zeros(2, depth, dtype=torch.float32, pin_memory=True)

        def _body():
            toks = torch.zeros(depth, dtype=torch.float32,
                               device=self.device)
            prbs = torch.zeros(depth, dtype=torch.float32,
                               device=self.device)
            cur = dev_in[0:1]
            pos0 = dev_in[1:2]
            for i in range(depth):
                h = self._embed(cur.view(1, 1))
                sp = (pos0 + i).to(torch.int32)
                h = self.stack.forward_inference(h, kv, sp)
                logits = self.logits_for(h[:, -1]).float()
                p = torch.softmax(logits[0], -1)
                pv, ti = p.max(-1)
                toks[i] = ti.to(torch.float32)
                prbs[i] = pv
                cur = ti.view(1)
            return torch.stack((toks, prbs))

        for _ in range(2):  # allocator/kernel warmup
            kv.extend(depth, speculative=True)
            kv.page_table()
            _body()
            kv.rollback()
        graph = torch.cuda.CUDAGraph()
        kv.extend(depth, speculative=True)
        kv.page_table()
        with torch.cuda.graph(graph, capture_error_mode="thread_local"):
            out_buf = _body()
        kv.rollback()

        def chain(tok: int, position0: int):
            staging[0] = tok
            staging[1] = position0
            dev_in.copy_(staging, non_blocking=True)
            kv.page_table()
            graph.replay()
            out_host.copy_(out_buf, non_blocking=False)  # syncs
            return ([int(t) for t in out_host[0]],
                    [float(p) for p in out_host[1]])

        return chain
