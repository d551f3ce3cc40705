"""Reachability checks (parity: reference server/reachability.py:22-164).

A joining server must be dialable by the swarm, or clients will route to it
and fail. `check_direct_reachability` asks an already-connected peer to dial
our listen address back over a fresh TCP connection (the reference's
ReachabilityProtocol `rpc_check`); `validate_reachability` retries with
a deadline. There is no centralized validator (the reference's
health.petals.dev) in this offline build — swarm peers are the probes.
"""

from __future__ import annotations

import asyncio
import logging
from typing import Optional, Sequence, Tuple

from petals_amd.p2p.transport import P2PNode, RpcError, RpcMessage

logger = logging.getLogger(__name__)

RPC_CHECK = "reachability.rpc_check"


class ReachabilityProtocol:
    """Serves dial-back requests: on rpc_check {host, port}, open a fresh TCP
    connection to that address and report success."""

    def __init__(self, p2p: P2PNode):
        self.p2p = p2p
        p2p.add_handler(RPC_CHECK, self.rpc_check)

    async def rpc_check(self, request: RpcMessage, stream) -> None:
        host = request.meta.get("host")
        port = int(request.meta.get("port", 0))
        ok = False
        if host and port:
            try:
                reader, writer = await asyncio.wait_for(asyncio.open_connection(host, port), 5.0)
                writer.close()
                ok = True
            except (OSError, asyncio.TimeoutError):
                ok = False
        await stream.close(RpcMessage(meta={"reachable": ok, "peer_id": self.p2p.peer_id}))


async def check_direct_reachability(
    p2p: P2PNode,
    own_addr: Tuple[str, int],
    probe_peers: Sequence[Tuple[str, int]],
) -> Optional[bool]:
    """Ask swarm peers to dial us back. Returns True/False, or None if no peer
    could answer (e.g. empty swarm)."""
    for addr in probe_peers:
        try:
            resp = await asyncio.wait_for(
                p2p.call_unary(
                    tuple(addr), RPC_CHECK, RpcMessage(meta={"host": own_addr[0], "port": own_addr[1]})
                ),
                timeout=10.0,
            )
            return bool(resp.meta.get("reachable"))
        except (RpcError, OSError, asyncio.TimeoutError) as e:
            logger.debug("reachability probe via %s failed: %r", addr, e)
    return None


async def validate_reachability(
    p2p: P2PNode,
    own_addr: Tuple[str, int],
    probe_peers: Sequence[Tuple[str, int]],
    *,
    wait_time: float = 60.0,
    retry_time: float = 10.0,
) -> None:
    """Raise if the swarm cannot dial us back within `wait_time` seconds."""
    deadline = asyncio.get_event_loop().time() + wait_time
    while True:
        ok = await check_direct_reachability(p2p, own_addr, probe_peers)
        if ok or ok is None:  # unreachable *probes* don't block a lone server
            if ok is None:
                logger.info("no peers could probe reachability (empty swarm?); continuing")
            return
        if asyncio.get_event_loop().time() > deadline:
            raise RuntimeError(
                f"server at {own_addr} is not reachable from the swarm. If behind NAT, "
                f"bind a public address/port or use a reachable relay host."
            )
        logger.warning("not yet reachable from the swarm; retrying in %.0f s", retry_time)
        await asyncio.sleep(retry_time)
