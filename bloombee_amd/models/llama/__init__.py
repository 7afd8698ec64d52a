from bloombee_amd.models.base import FamilyEntry, register_model_family
from bloombee_amd.models.llama.block import LlamaBlock, RopeTables  # noqa: F401
from bloombee_amd.models.llama.config import LLAMA_PRESETS, LlamaConfig  # noqa: F401

register_model_family(
    "llama",
    FamilyEntry(config_cls=LlamaConfig, block_cls=LlamaBlock, presets=LLAMA_PRESETS),
)
