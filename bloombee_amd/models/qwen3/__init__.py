from bloombee_amd.models.base import FamilyEntry, register_model_family
from bloombee_amd.models.llama.model import (DistributedLlamaForCausalLM,
                                             DistributedLlamaModel)
from bloombee_amd.models.qwen3.block import Qwen3Block  # noqa: F401
from bloombee_amd.models.qwen3.config import QWEN3_PRESETS, Qwen3Config  # noqa: F401

# Qwen3 shares the llama-style client head (rms final norm; large 152k vocab
# => GPU client recommended, ref README.md:99)
register_model_family(
    "qwen3",
    FamilyEntry(config_cls=Qwen3Config, block_cls=Qwen3Block,
                model_cls=DistributedLlamaModel,
                causal_lm_cls=DistributedLlamaForCausalLM,
                presets=QWEN3_PRESETS),
)
