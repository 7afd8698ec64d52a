from bloombee_amd.models.base import (  # noqa: F401
    FamilyEntry,
    ModelConfig,
    get_family,
    register_model_family,
    resolve_config,
)

# Import families for registration side effects.
import bloombee_amd.models.llama  # noqa: F401,E402
import bloombee_amd.models.bloom  # noqa: F401,E402
import bloombee_amd.models.falcon  # noqa: F401,E402
import bloombee_amd.models.qwen3  # noqa: F401,E402
import bloombee_amd.models.mixtral  # noqa: F401,E402
import bloombee_amd.models.gemma4  # noqa: F401,E402
