from bloombee_amd.models.base import FamilyEntry, register_model_family
from bloombee_amd.models.gemma4.block import Gemma4Block  # noqa: F401
from bloombee_amd.models.gemma4.config import GEMMA4_PRESETS, Gemma4Config  # noqa: F401
from bloombee_amd.models.gemma4.model import (  # noqa: F401
    DistributedGemma4ForCausalLM,
    DistributedGemma4Model,
)

register_model_family(
    "gemma4",
    FamilyEntry(config_cls=Gemma4Config, block_cls=Gemma4Block,
                model_cls=DistributedGemma4Model,
                causal_lm_cls=DistributedGemma4ForCausalLM,
                presets=GEMMA4_PRESETS),
)
