"""bloombee_amd — an MI355X-native decentralized LLM serving & fine-tuning engine.

Capabilities modeled on ai-decentralized/BloomBee (see /root/repo/SURVEY.md):
a client holds embeddings + LM head and routes hidden states through a chain
of workers, each serving a contiguous range of transformer blocks. On a single
8xMI355X node the inter-worker hops run as RCCL send/recv over xGMI; every hot
transformer-block primitive is a hand-written CDNA4 (gfx950) HIP kernel.

This package is a from-scratch design, not a port: no CUDA shims, no Triton,
no hipify output.
"""

__version__ = "0.1.0"

from bloombee_amd.models.auto import (  # noqa: F401
    AutoDistributedConfig,
    AutoDistributedModel,
    AutoDistributedModelForCausalLM,
    AutoDistributedModelForSequenceClassification,
    AutoDistributedSpeculativeModel,
)
