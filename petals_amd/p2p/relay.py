"""Circuit relay: lets a NAT'd/unreachable server serve through a reachable
peer (the reference got this from libp2p's circuit relays + hole punching,
reference server/server.py:137-150, reachability.py:86-164).

Design: the relay is a rendezvous + byte pump — it NEVER terminates the
session protocol. An unreachable server S keeps a framed `relay.register`
control stream open to relay R. A client dials R with a raw `PAMDR` preamble
naming S's peer id; R asks S (over the control stream) to dial back a raw
`PAMDT` tunnel with a one-time token, then splices the two TCP streams byte
for byte. The client then runs the NORMAL transport handshake (magic +
optional STARTTLS) end-to-end through the splice, so with secure mode on the
relay cannot read or tamper with the session (it only learns the metadata:
who talks to whom).

Announcements: S announces `addr = [relay_host, relay_port, "relay",
S.peer_id]` and ServerInfo.using_relay = True; routing applies the reference's
x0.2 relay throughput penalty (reference server/throughput.py:96-107).
"""

from __future__ import annotations

import asyncio
import logging
import os
from typing import Dict, Tuple

from petals_amd.p2p.transport import P2PNode, RpcError, RpcMessage, RpcStream

logger = logging.getLogger(__name__)

TUNNEL_TIMEOUT = 20.0


async def _pump(reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
    try:
        while True:
            data = await reader.read(1 << 16)
            if not data:
                break
            writer.write(data)
            await writer.drain()
    except Exception:  # noqa: BLE001
        pass
    finally:
        try:
            writer.close()
        except Exception:  # noqa: BLE001
            pass


class RelayHub:
    """Runs on a reachable server's P2PNode: rendezvous + splice."""

    def __init__(self, p2p: P2PNode):
        self.p2p = p2p
        self.registered: Dict[str, RpcStream] = {}
        self.pending: Dict[str, asyncio.Future] = {}
        p2p.add_handler("relay.register", self._handle_register)
        p2p.relay_hub = self

    async def _handle_register(self, request: RpcMessage, stream: RpcStream) -> None:
        peer_id = request.meta.get("peer_id")
        if not peer_id:
            raise RpcError("relay.register needs a peer_id")
        self.registered[peer_id] = stream
        logger.info("relay: registered peer %s", peer_id[:8])
        try:
            while True:  # hold the control stream open until the peer leaves
                await stream.receive(timeout=None)
        except RpcError:
            pass
        finally:
            if self.registered.get(peer_id) is stream:
                del self.registered[peer_id]
            logger.info("relay: peer %s gone", peer_id[:8])

    async def open_tunnel(self, target: str, reader: asyncio.StreamReader,
                          writer: asyncio.StreamWriter) -> None:
        """Client leg arrived (PAMDR): rendezvous with the target and splice."""
        control = self.registered.get(target)
        if control is None:
            writer.write(b"NO\n")
            await writer.drain()
            writer.close()
            return
        token = os.urandom(8).hex()
        fut: asyncio.Future = asyncio.get_event_loop().create_future()
        self.pending[token] = fut
        try:
            await control.send(RpcMessage(meta={"token": token}))
            t_reader, t_writer = await asyncio.wait_for(fut, TUNNEL_TIMEOUT)
        except Exception:  # noqa: BLE001
            self.pending.pop(token, None)
            writer.write(b"NO\n")
            await writer.drain()
            writer.close()
            return
        writer.write(b"GO\n")
        t_writer.write(b"GO\n")
        await writer.drain()
        await t_writer.drain()
        asyncio.ensure_future(_pump(reader, t_writer))
        asyncio.ensure_future(_pump(t_reader, writer))

    def accept_tunnel(self, token: str, reader, writer) -> bool:
        """Target leg arrived (PAMDT)."""
        fut = self.pending.pop(token, None)
        if fut is None or fut.done():
            return False
        fut.set_result((reader, writer))
        return True


class RelayClient:
    """Runs on the UNREACHABLE server: keeps the control stream open and
    dials back tunnels on demand."""

    def __init__(self, p2p: P2PNode, relay_addr: Tuple[str, int]):
        self.p2p = p2p
        self.relay_addr = (relay_addr[0], int(relay_addr[1]))
        self._task = None

    async def start(self) -> None:
        stream = await self.p2p.open_stream(
            self.relay_addr, "relay.register", RpcMessage(meta={"peer_id": self.p2p.peer_id})
        )
        self._task = asyncio.ensure_future(self._listen(stream))

    async def _listen(self, stream: RpcStream) -> None:
        try:
            while True:
                msg = await stream.receive(timeout=None)
                token = msg.meta.get("token")
                if token:
                    asyncio.ensure_future(self._dial_tunnel(token))
        except RpcError as e:
            logger.warning("relay control stream closed: %r", e)

    async def _dial_tunnel(self, token: str) -> None:
        try:
            reader, writer = await asyncio.open_connection(*self.relay_addr)
            writer.write(b"PAMDT\n" + token.encode() + b"\n")
            await writer.drain()
            go = await asyncio.wait_for(reader.readexactly(3), TUNNEL_TIMEOUT)
            if go != b"GO\n":
                writer.close()
                return
            # from here the splice is live: act as the ACCEPTING side of a
            # normal transport handshake (magic + optional STARTTLS)
            await self.p2p.accept_stream_pair(reader, writer)
        except Exception as e:  # noqa: BLE001
            logger.warning("relay tunnel dial-back failed: %r", e)


async def connect_via_relay(p2p: P2PNode, relay_addr: Tuple[str, int], target_peer: str,
                            timeout: float) -> Tuple[asyncio.StreamReader, asyncio.StreamWriter]:
    """Client side: returns a raw stream pair spliced through the relay to the
    target; the caller runs the normal outbound handshake over it."""
    reader, writer = await asyncio.wait_for(asyncio.open_connection(*relay_addr), timeout)
    writer.write(b"PAMDR\n" + target_peer.encode() + b"\n")
    await writer.drain()
    resp = await asyncio.wait_for(reader.readexactly(3), TUNNEL_TIMEOUT)
    if resp != b"GO\n":
        writer.close()
        raise RpcError(f"relay at {relay_addr} could not reach peer {target_peer[:8]}")
    return reader, writer
