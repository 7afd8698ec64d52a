"""Reachability validation (parity: reference server/reachability.py:20-171 —
can peers actually connect to our announced endpoint? The reference probes
via a health API + libp2p protocol; here a peer-assisted RPC probe)."""
from __future__ import annotations

import asyncio
from typing import Optional, Sequence, Tuple

from bloombee_amd.net.rpc import RpcClient, RpcServer
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


def attach_reachability(server: RpcServer) -> None:
    """Adds rpc_check_reachability: the callee dials BACK the address the
    caller claims to serve on and reports success (ReachabilityProtocol)."""

    async def check(meta, tensors):
        host, port = meta["host"], int(meta["port"])
        try:
            client = RpcClient(host, port)
            await client.call("rpc_info", {}, timeout=5)
            await client.close()
            return {"reachable": True}, []
        except Exception as e:  # noqa: BLE001
            return {"reachable": False, "error": str(e)}, []

    server.register("rpc_check_reachability", check)


async def check_direct_reachability(my_endpoint: Tuple[str, int],
                                    via_peers: Sequence[Tuple[str, int]],
                                    ) -> Optional[bool]:
    """Ask peers to dial us back; True if any succeeds, False if all report
    failure, None if nobody answered (ref check_direct_reachability)."""
    answered = False
    for peer in via_peers:
        try:
            client = RpcClient(*peer)
            meta, _ = await client.call(
                "rpc_check_reachability",
                {"host": my_endpoint[0], "port": my_endpoint[1]}, timeout=10)
            await client.close()
            answered = True
            if meta.get("reachable"):
                return True
        except Exception:  # noqa: BLE001
            continue
    return False if answered else None
