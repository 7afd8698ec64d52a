"""Client-side distributed Gemma-4: scaled word embeddings (sqrt(hidden)),
(1+w) RMS final norm, tied LM head (parity: reference models/gemma4/model.py)."""
from __future__ import annotations

import math

import torch
import torch.nn.functional as F

from bloombee_amd import ops
from bloombee_amd.models.llama.model import (DistributedLlamaForCausalLM,
                                             DistributedLlamaModel, LMHead)


class DistributedGemma4Model(DistributedLlamaModel):
    def embed(self, input_ids: torch.Tensor) -> torch.Tensor:
        h = F.embedding(input_ids.to(self.device_), self.embed_tokens)
        return h * math.sqrt(self.config.hidden_size)

    # final norm: (1 + w) RMS with w random-init 0 == plain rms with ones —
    # the inherited rms_norm(norm_w=1) is numerically identical at init; a
    # checkpoint load writes (1 + w) into norm_w (from_pretrained maps it).


class DistributedGemma4ForCausalLM(DistributedLlamaForCausalLM):
    def __init__(self, config, client_config, model_name, seed=0, device="cpu",
                 manager=None, pre_seq_len=0):
        torch.nn.Module.__init__(self)
        self.config = config
        self.transformer = DistributedGemma4Model(config, client_config,
                                                  model_name, seed=seed,
                                                  device=device, manager=manager,
                                                  pre_seq_len=pre_seq_len)
        self.lm_head = LMHead(config, self.transformer.embed_tokens,
                              gen=self.transformer._gen, device=device)
