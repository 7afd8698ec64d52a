from bloombee_amd.kv.paged import (  # noqa: F401
    AllocationFailed,
    PagedKVCache,
    PagedKVError,
    SessionHandle,
)
