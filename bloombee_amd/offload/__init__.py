from bloombee_amd.offload.policy import OffloadPolicy  # noqa: F401
from bloombee_amd.offload.weights import OffloadedBlockStack  # noqa: F401
