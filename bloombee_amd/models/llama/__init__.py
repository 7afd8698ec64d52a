from bloombee_amd.models.base import FamilyEntry, register_model_family
from bloombee_amd.models.llama.block import LlamaBlock, RopeTables  # noqa: F401
from bloombee_amd.models.llama.config import LLAMA_PRESETS, LlamaConfig  # noqa: F401


def _entry():
    # model classes import the client stack lazily to avoid a hard cycle at
    # package import time
    from bloombee_amd.models.llama.model import (
        DistributedLlamaForCausalLM, DistributedLlamaForSequenceClassification,
        DistributedLlamaModel)
    from bloombee_amd.models.llama.speculative import \
        DistributedLlamaForSpeculativeGeneration
    return (DistributedLlamaModel, DistributedLlamaForCausalLM,
            DistributedLlamaForSpeculativeGeneration,
            DistributedLlamaForSequenceClassification)


try:
    _model_cls, _causal_cls, _spec_cls, _seq_cls = _entry()
except ImportError:  # pragma: no cover
    _model_cls = _causal_cls = _spec_cls = _seq_cls = None

register_model_family(
    "llama",
    FamilyEntry(config_cls=LlamaConfig, block_cls=LlamaBlock,
                model_cls=_model_cls, causal_lm_cls=_causal_cls,
                speculative_cls=_spec_cls, seq_cls_cls=_seq_cls,
                presets=LLAMA_PRESETS),
)
