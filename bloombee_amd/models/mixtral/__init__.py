from bloombee_amd.models.base import FamilyEntry, register_model_family
from bloombee_amd.models.llama.model import (DistributedLlamaForCausalLM,
                                             DistributedLlamaModel)
from bloombee_amd.models.mixtral.block import MixtralBlock  # noqa: F401
from bloombee_amd.models.mixtral.config import (  # noqa: F401
    MIXTRAL_PRESETS,
    MixtralConfig,
)

register_model_family(
    "mixtral",
    FamilyEntry(config_cls=MixtralConfig, block_cls=MixtralBlock,
                model_cls=DistributedLlamaModel,
                causal_lm_cls=DistributedLlamaForCausalLM,
                presets=MIXTRAL_PRESETS),
)
