from bloombee_amd.client.config import ClientConfig  # noqa: F401
from bloombee_amd.client.routing import RemoteSequenceManager  # noqa: F401
from bloombee_amd.client.session import InferenceSession  # noqa: F401
from bloombee_amd.client.sequential import RemoteSequential  # noqa: F401
