"""Client-side distributed BLOOM (parity: reference models/bloom/model.py:
DistributedBloomModel/ForCausalLM — embeddings + word_embeddings_layernorm on
the client, LN final head, tied LM head)."""
from __future__ import annotations

import torch
import torch.nn.functional as F

from bloombee_amd import ops
from bloombee_amd.models.llama.model import (DistributedLlamaForCausalLM,
                                             DistributedLlamaModel)


class DistributedBloomModel(DistributedLlamaModel):
    def __init__(self, *args, **kw):
        super().__init__(*args, **kw)
        cfg = self.config
        dt = cfg.dtype
        dev = self.device_
        self.embed_ln_w = torch.nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=dt, device=dev), requires_grad=False)
        self.embed_ln_b = torch.nn.Parameter(
            torch.zeros(cfg.hidden_size, dtype=dt, device=dev), requires_grad=False)
        self.norm_b = torch.nn.Parameter(
            torch.zeros(cfg.hidden_size, dtype=dt, device=dev), requires_grad=False)

    def embed(self, input_ids: torch.Tensor) -> torch.Tensor:
        h = F.embedding(input_ids.to(self.device_), self.embed_tokens)
        return ops.layer_norm(h, self.embed_ln_w, self.embed_ln_b,
                              self.config.layer_norm_epsilon)

    def final_norm(self, hidden: torch.Tensor) -> torch.Tensor:
        return ops.layer_norm(hidden, self.norm_w, self.norm_b,
                              self.config.layer_norm_epsilon)


class DistributedBloomForCausalLM(DistributedLlamaForCausalLM):
    _model_cls = DistributedBloomModel

    def __init__(self, config, client_config, model_name, seed=0, device="cpu",
                 manager=None, pre_seq_len=0):
        torch.nn.Module.__init__(self)
        self.config = config
        self.transformer = DistributedBloomModel(config, client_config,
                                                 model_name, seed=seed,
                                                 device=device, manager=manager,
                                                 pre_seq_len=pre_seq_len)
        from bloombee_amd.models.llama.model import LMHead
        self.lm_head = LMHead(config, self.transformer.embed_tokens,
                              gen=self.transformer._gen, device=device)
