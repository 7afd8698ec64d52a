"""Generation over the swarm (parity: reference RemoteGenerationMixin,
client/remote_generation.py:141-386 — greedy and sampling loops over an
InferenceSession, session reuse across calls)."""
from __future__ import annotations

from typing import Optional

import torch

from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


class RemoteGenerationMixin:
    """Requires: self.transformer (with .remote embedding/norm API below) and
    self.lm_head. The host model implements:
        embed(input_ids) -> hidden
        final_norm(hidden) -> hidden
        lm_head(hidden) -> logits
        remote (RemoteSequential)
    """

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 20,
                 do_sample: bool = False, temperature: float = 1.0,
                 top_k: Optional[int] = None, top_p: Optional[float] = None,
                 session=None,
                 generator: Optional[torch.Generator] = None) -> torch.Tensor:
        B, T = input_ids.shape
        # p-tuning at inference (ref remote_generation + ptune): trained
        # prompt embeds are prepended once at prefill; deep per-block prompts
        # ride the first step's item and are re-applied by every span
        pe = getattr(self.transformer, "prompt_embeds", None) \
            if hasattr(self, "transformer") else None
        dp = getattr(self.transformer, "deep_prompts", None) \
            if hasattr(self, "transformer") else None
        pre = pe.shape[0] if pe is not None else 0
        own_session = session is None
        if own_session:
            session = self.remote.inference_session(
                T + pre + max_new_tokens + 1)
        out_tokens = [input_ids]
        try:
            hidden = self.embed(input_ids)
            if pe is not None and session.position == 0:
                p = pe.detach().to(hidden.dtype).unsqueeze(0).expand(B, -1, -1)
                hidden = torch.cat([p, hidden], dim=1)
            hidden = session.step(
                hidden, prompts=dp.detach() if dp is not None else None)
            logits = self.lm_head(self.final_norm(hidden[:, -1:]))[:, -1]
            next_tok = self._pick(logits, do_sample, temperature, top_k,
                                  top_p, generator)
            out_tokens.append(next_tok.view(B, 1))
            for _ in range(max_new_tokens - 1):
                hidden = self.embed(next_tok.view(B, 1))
                hidden = session.step(hidden)
                logits = self.lm_head(self.final_norm(hidden[:, -1:]))[:, -1]
                next_tok = self._pick(logits, do_sample, temperature, top_k,
                                  top_p, generator)
                out_tokens.append(next_tok.view(B, 1))
        finally:
            if own_session:
                session.close()
        return torch.cat(out_tokens, dim=1)

    @staticmethod
    def _pick(logits: torch.Tensor, do_sample: bool, temperature: float,
              top_k: Optional[int], top_p: Optional[float],
              generator: Optional[torch.Generator] = None) -> torch.Tensor:
        if not do_sample:
            return logits.argmax(dim=-1)
        logits = logits.float() / max(temperature, 1e-6)
        if top_k is not None:
            kth = logits.topk(top_k, dim=-1).values[..., -1:]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        if top_p is not None:
            sorted_logits, idx = logits.sort(dim=-1, descending=True)
            probs = sorted_logits.softmax(-1).cumsum(-1)
            kill = probs - probs.new_zeros(probs.shape).scatter_(
                -1, torch.zeros_like(idx[..., :1]), 0) > top_p
            kill[..., 0] = False
            mask = torch.zeros_like(logits, dtype=torch.bool).scatter_(
                -1, idx, kill)
            logits = logits.masked_fill(mask, float("-inf"))
        return torch.multinomial(logits.softmax(-1), 1,
                                 generator=generator).squeeze(-1)
