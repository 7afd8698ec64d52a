"""Client-side distributed Falcon: LN final head, tied embeddings (parity:
reference models/falcon/model.py)."""
from __future__ import annotations

import torch

from bloombee_amd import ops
from bloombee_amd.models.llama.model import (DistributedLlamaForCausalLM,
                                             DistributedLlamaModel, LMHead)


class DistributedFalconModel(DistributedLlamaModel):
    def __init__(self, *args, **kw):
        super().__init__(*args, **kw)
        self.norm_b = torch.nn.Parameter(
            torch.zeros(self.config.hidden_size, dtype=self.config.dtype,
                        device=self.device_), requires_grad=False)

    def final_norm(self, hidden: torch.Tensor) -> torch.Tensor:
        return ops.layer_norm(hidden, self.norm_w, self.norm_b,
                              self.config.layer_norm_epsilon)


class DistributedFalconForCausalLM(DistributedLlamaForCausalLM):
    def __init__(self, config, client_config, model_name, seed=0, device="cpu",
                 manager=None, pre_seq_len=0):
        torch.nn.Module.__init__(self)
        self.config = config
        self.transformer = DistributedFalconModel(config, client_config,
                                                  model_name, seed=seed,
                                                  device=device, manager=manager,
                                                  pre_seq_len=pre_seq_len)
        self.lm_head = LMHead(config, self.transformer.embed_tokens,
                              gen=self.transformer._gen, device=device)
