from bloombee_amd.models.base import FamilyEntry, register_model_family
from bloombee_amd.models.bloom.block import BloomBlock  # noqa: F401
from bloombee_amd.models.bloom.config import BLOOM_PRESETS, BloomConfig  # noqa: F401
from bloombee_amd.models.bloom.model import (  # noqa: F401
    DistributedBloomForCausalLM,
    DistributedBloomModel,
)

register_model_family(
    "bloom",
    FamilyEntry(config_cls=BloomConfig, block_cls=BloomBlock,
                model_cls=DistributedBloomModel,
                causal_lm_cls=DistributedBloomForCausalLM,
                presets=BLOOM_PRESETS),
)
