"""Client-side distributed LLaMA (parity: reference models/llama/model.py:
DistributedLlamaModel / DistributedLlamaForCausalLM — the client holds
embeddings, final norm and LM head; the block stack runs on the swarm via
RemoteSequential; p-tuning prompts stay fp32 on the client, ref
client/ptune.py)."""
from __future__ import annotations

import math
from typing import Optional, Sequence, Tuple

import torch
import torch.nn.functional as F

from bloombee_amd import ops
from bloombee_amd.client.config import ClientConfig
from bloombee_amd.client.generation import RemoteGenerationMixin
from bloombee_amd.client.routing import RemoteSequenceManager
from bloombee_amd.client.sequential import RemoteSequential
from bloombee_amd.models.base import ModelConfig, resolve_config


class DistributedLlamaModel(torch.nn.Module):
    """Embeddings + remote block chain + final norm."""

    def __init__(self, config: ModelConfig, client_config: ClientConfig,
                 model_name: str, seed: int = 0, device: str = "cpu",
                 manager: Optional[RemoteSequenceManager] = None,
                 pre_seq_len: int = 0,
                 deep_ptune: bool = False,
                 checkpoint_dir: Optional[str] = None):
        super().__init__()
        self.config = config
        self.device_ = torch.device(device)
        # one generator, embed drawn first then (untied) lm_head — the SAME
        # draw order as LocalEngine so random-init swarm == random-init local
        gen = torch.Generator().manual_seed(seed)
        self._gen = gen
        dt = config.dtype
        self.embed_tokens = torch.nn.Parameter(
            torch.randn(config.vocab_size, config.hidden_size, generator=gen)
            .mul_(0.02).to(dt).to(device), requires_grad=False)
        self.norm_w = torch.nn.Parameter(
            torch.ones(config.hidden_size, dtype=dt, device=device),
            requires_grad=False)
        # real weights: the client's share of a converted checkpoint
        # (embed + final norm; the LM head loads its own file)
        self.checkpoint_dir = checkpoint_dir
        if checkpoint_dir is not None:
            from bloombee_amd.server.from_pretrained import load_client_weights
            cw = load_client_weights(checkpoint_dir)
            if cw["embed"] is None or cw["final_norm"] is None:
                raise FileNotFoundError(
                    f"no client weights (embed_tokens/norm) under "
                    f"{checkpoint_dir}")
            with torch.no_grad():
                self.embed_tokens.copy_(cw["embed"].to(dt))
                self.norm_w.copy_(cw["final_norm"].to(dt))
        self.remote = RemoteSequential(client_config, model_name,
                                       config.num_hidden_layers,
                                       manager=manager)
        # p-tuning: trainable prompt embeddings kept fp32 on the client
        # (ref client/ptune.py:23-41). deep_ptune additionally trains one
        # additive prompt per transformer block (ref ptune.py deep mode:
        # intermediate_prompts added to the prompt positions at every
        # block input; grads ride back on rpc_backward).
        self.pre_seq_len = pre_seq_len
        if pre_seq_len > 0:
            self.prompt_embeds = torch.nn.Parameter(
                torch.randn(pre_seq_len, config.hidden_size, device=device)
                .mul_(0.02))
        else:
            self.prompt_embeds = None
        if pre_seq_len > 0 and deep_ptune:
            self.deep_prompts = torch.nn.Parameter(
                torch.zeros(config.num_hidden_layers, pre_seq_len,
                            config.hidden_size, device=device))
        else:
            self.deep_prompts = None

    def embed(self, input_ids: torch.Tensor) -> torch.Tensor:
        h = F.embedding(input_ids.to(self.device_), self.embed_tokens)
        return h

    def final_norm(self, hidden: torch.Tensor) -> torch.Tensor:
        return ops.rms_norm(hidden, self.norm_w, self.config.rms_norm_eps)

    def trainable_parameters(self):
        """The client-side fine-tune state (ref ptune.py: everything the
        swarm trains lives on the client — shallow + deep prompts; server
        blocks stay frozen). Feed to a torch optimizer directly."""
        ps = []
        if self.prompt_embeds is not None:
            ps.append(self.prompt_embeds)
        if self.deep_prompts is not None:
            ps.append(self.deep_prompts)
        return ps

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        """Training-path full forward (no KV): returns final hidden states."""
        h = self.embed(input_ids)
        if self.prompt_embeds is not None:
            B = h.shape[0]
            p = self.prompt_embeds.to(h.dtype).unsqueeze(0).expand(B, -1, -1)
            h = torch.cat([p, h], dim=1)
        h = self.remote(h, prompts=self.deep_prompts)
        return self.final_norm(h)


class LMHead(torch.nn.Module):
    """Client-side vocab projection (ref client/lm_head.py)."""

    def __init__(self, config: ModelConfig, embed: torch.nn.Parameter,
                 gen: torch.Generator, device: str = "cpu",
                 checkpoint_dir: Optional[str] = None):
        super().__init__()
        if config.tie_word_embeddings:
            self.weight = embed
        else:
            self.weight = torch.nn.Parameter(
                torch.randn(config.vocab_size, config.hidden_size, generator=gen)
                .mul_(0.02).to(config.dtype).to(device), requires_grad=False)
            if checkpoint_dir is not None:
                from bloombee_amd.server.from_pretrained import \
                    load_client_weights
                w = load_client_weights(checkpoint_dir)["lm_head"]
                if w is None:
                    raise FileNotFoundError(
                        f"untied model but no lm_head.weight under "
                        f"{checkpoint_dir}")
                with torch.no_grad():
                    self.weight.copy_(w.to(self.weight.dtype))

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        return ops.linear(hidden, self.weight)


class DistributedLlamaForCausalLM(RemoteGenerationMixin, torch.nn.Module):
    def __init__(self, config: ModelConfig, client_config: ClientConfig,
                 model_name: str, seed: int = 0, device: str = "cpu",
                 manager: Optional[RemoteSequenceManager] = None,
                 pre_seq_len: int = 0,
                 deep_ptune: bool = False,
                 checkpoint_dir: Optional[str] = None):
        super().__init__()
        self.config = config
        self.transformer = DistributedLlamaModel(config, client_config,
                                                 model_name, seed=seed,
                                                 device=device, manager=manager,
                                                 pre_seq_len=pre_seq_len,
                                                 deep_ptune=deep_ptune,
                                                 checkpoint_dir=checkpoint_dir)
        self.lm_head = LMHead(config, self.transformer.embed_tokens,
                              gen=self.transformer._gen, device=device,
                              checkpoint_dir=checkpoint_dir)

    # RemoteGenerationMixin hooks
    @property
    def remote(self) -> RemoteSequential:
        return self.transformer.remote

    def embed(self, input_ids):
        return self.transformer.embed(input_ids)

    def final_norm(self, hidden):
        return self.transformer.final_norm(hidden)

    def trainable_parameters(self):
        return self.transformer.trainable_parameters()

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        """Training-path logits (differentiable wrt prompts)."""
        h = self.transformer(input_ids)
        return self.lm_head(h)

    @classmethod
    def from_pretrained(cls, name_or_path: str, config: Optional[ModelConfig] = None,
                        client_config: Optional[ClientConfig] = None,
                        initial_peers: Sequence[Tuple[str, int]] = (),
                        seed: int = 0, device: str = "cpu", **kw):
        cfg = config or resolve_config(name_or_path)
        ccfg = client_config or ClientConfig(initial_peers=list(initial_peers))
        # a converted checkpoint directory serves real client weights
        # (ref client/from_pretrained.py: the client loads only its share)
        if "checkpoint_dir" not in kw:
            from bloombee_amd.server.from_pretrained import is_converted
            if is_converted(name_or_path):
                kw["checkpoint_dir"] = name_or_path
        return cls(cfg, ccfg, model_name=name_or_path, seed=seed,
                   device=device, **kw)


class DistributedLlamaForSequenceClassification(torch.nn.Module):
    """Classification over the swarm (parity: reference
    AutoDistributedModelForSequenceClassification — the SST-2 prompt-tuning
    notebook's model class): frozen remote blocks + a TRAINABLE client-side
    score head over the last token's hidden state; combine with
    pre_seq_len/deep_ptune for prompt-tuned classification."""

    def __init__(self, config: ModelConfig, client_config: ClientConfig,
                 model_name: str, num_labels: int = 2, seed: int = 0,
                 device: str = "cpu",
                 manager: Optional[RemoteSequenceManager] = None,
                 pre_seq_len: int = 0, deep_ptune: bool = False):
        super().__init__()
        self.config = config
        self.num_labels = num_labels
        self.transformer = DistributedLlamaModel(config, client_config,
                                                 model_name, seed=seed,
                                                 device=device,
                                                 manager=manager,
                                                 pre_seq_len=pre_seq_len,
                                                 deep_ptune=deep_ptune)
        gen = torch.Generator().manual_seed(seed + 7)
        self.score = torch.nn.Parameter(
            torch.randn(num_labels, config.hidden_size, generator=gen)
            .mul_(0.02).float().to(device))

    @property
    def remote(self) -> RemoteSequential:
        return self.transformer.remote

    def trainable_parameters(self):
        return self.transformer.trainable_parameters() + [self.score]

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        """-> (B, num_labels) logits from the LAST token's hidden state."""
        h = self.transformer(input_ids)
        return F.linear(h[:, -1].float(), self.score)

    @classmethod
    def from_pretrained(cls, name_or_path: str,
                        config: Optional[ModelConfig] = None,
                        client_config: Optional[ClientConfig] = None,
                        initial_peers: Sequence[Tuple[str, int]] = (),
                        seed: int = 0, device: str = "cpu", **kw):
        cfg = config or resolve_config(name_or_path)
        ccfg = client_config or ClientConfig(initial_peers=list(initial_peers))
        return cls(cfg, ccfg, model_name=name_or_path, seed=seed,
                   device=device, **kw)
