"""Networking: tensor wire codec, asyncio RPC, and the DHT control plane.

Replaces the reference's hivemind/libp2p dependency stack (SURVEY.md §2.7,
§5 "Distributed communication backend"): the control plane (discovery,
announcements) is a Kademlia-style DHT over our own asyncio RPC; the data
plane is the same RPC for off-node hops and RCCL/xGMI channels for same-node
GPU→GPU pushes (parallel/xgmi.py).
"""
from bloombee_amd.net.tensors import (  # noqa: F401
    deserialize_tensor,
    pack_frame,
    serialize_tensor,
    unpack_frame,
)
from bloombee_amd.net.rpc import RpcClient, RpcError, RpcServer  # noqa: F401
