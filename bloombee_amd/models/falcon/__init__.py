from bloombee_amd.models.base import FamilyEntry, register_model_family
from bloombee_amd.models.falcon.block import FalconBlock  # noqa: F401
from bloombee_amd.models.falcon.config import FALCON_PRESETS, FalconConfig  # noqa: F401
from bloombee_amd.models.falcon.model import (  # noqa: F401
    DistributedFalconForCausalLM,
    DistributedFalconModel,
)

register_model_family(
    "falcon",
    FamilyEntry(config_cls=FalconConfig, block_cls=FalconBlock,
                model_cls=DistributedFalconModel,
                causal_lm_cls=DistributedFalconForCausalLM,
                presets=FALCON_PRESETS),
)
