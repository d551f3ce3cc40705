"""Framed-msgpack TCP transport with unary and bidirectional-stream RPCs.

This replaces the reference's go-libp2p daemon (`hivemind.P2P`, spawned per
process — see SURVEY §2.3) with an in-process asyncio transport. The wire unit
is a *frame*:

    [u32 big-endian length][msgpack map]

and the msgpack map is an *envelope*:

    {"rpc": str, "rid": int, "kind": "req"|"item"|"end"|"err",
     "meta": {...},                      # msgpack-able metadata side-channel
     "tdescs": [tensor descriptors],     # see utils.serialization
     "tbufs": [bytes, ...]}              # raw tensor payloads

Every RPC is a message stream keyed by `rid`: a unary call is a stream with one
"req" inbound and one "end" outbound. Handlers receive an `RpcStream` with an
async-iterator of inbound messages and `send()`/`close()` for outbound ones.

Identity & encryption: by default a node's peer id is a random 16-byte hex
string and traffic is plaintext (fastest for trusted fabrics). With
``secure=True`` the node owns an ed25519 keypair (p2p/identity.py), its peer
id is the certificate fingerprint, and every connection starts with a magic
line then upgrades via STARTTLS (TLS 1.3): clients verify that the server's
certificate fingerprint equals the peer id announced in the DHT, so a relay
or on-path attacker cannot impersonate a server. (The reference got identity
+ encryption + relays from go-libp2p; see also relay.py for the circuit-relay
fallback for NAT'd servers.)

Wire handshake (dialer speaks first): b"PAMD1\n" = plaintext frames follow;
b"PAMDS\n" = server replies b"OK\n" and both sides wrap in TLS, then frames.
"""

from __future__ import annotations

import asyncio
import logging
import os
import socket
import struct
from typing import Any, Awaitable, Callable, Dict, List, Optional, Tuple

import msgpack
import torch

from petals_amd.utils import serialization

logger = logging.getLogger(__name__)

MAX_FRAME_SIZE = 1 << 30  # 1 GiB hard cap per frame
# payloads larger than this are split into multiple stream items by callers
MAX_UNARY_PAYLOAD_SIZE = 64 << 20


class RpcError(Exception):
    pass


class RpcMessage:
    """One envelope: metadata + a list of tensors."""

    __slots__ = ("meta", "tensors", "kind")

    def __init__(self, meta: Optional[Dict[str, Any]] = None, tensors: Optional[List[torch.Tensor]] = None, kind: str = "item"):
        self.meta = meta or {}
        self.tensors = tensors or []
        self.kind = kind

    def __repr__(self):
        return f"RpcMessage(kind={self.kind}, meta_keys={list(self.meta)}, n_tensors={len(self.tensors)})"


def _encode_envelope(rpc: str, rid: int, kind: str, msg: RpcMessage, compressions=None) -> bytes:
    tdescs, tbufs = serialization.serialize_tensors(msg.tensors, compressions)
    body = msgpack.packb(
        {"rpc": rpc, "rid": rid, "kind": kind, "meta": msg.meta, "tdescs": tdescs, "tbufs": tbufs},
        use_bin_type=True,
    )
    return struct.pack(">I", len(body)) + body


def _decode_envelope(body: bytes) -> Tuple[str, int, str, RpcMessage]:
    obj = msgpack.unpackb(body, raw=False, strict_map_key=False)
    tensors = serialization.deserialize_tensors(obj.get("tdescs", []), obj.get("tbufs", []))
    msg = RpcMessage(meta=obj.get("meta", {}), tensors=tensors, kind=obj["kind"])
    return obj["rpc"], obj["rid"], obj["kind"], msg


class RpcStream:
    """Bidirectional message stream for one RPC invocation."""

    def __init__(self, conn: "Connection", rpc: str, rid: int):
        self._conn = conn
        self.rpc = rpc
        self.rid = rid
        self._inbox: asyncio.Queue = asyncio.Queue()
        self._closed_outbound = False
        self._finished_inbound = False

    async def send(self, msg: RpcMessage, kind: str = "item", compressions=None) -> None:
        if self._closed_outbound:
            raise RpcError(f"stream {self.rpc}#{self.rid} already closed for sending")
        if kind == "end":
            self._closed_outbound = True
        await self._conn.send_frame(_encode_envelope(self.rpc, self.rid, kind, msg, compressions))

    async def close(self, msg: Optional[RpcMessage] = None) -> None:
        """Send the final message (or an empty end marker)."""
        if not self._closed_outbound:
            await self.send(msg or RpcMessage(), kind="end")

    async def error(self, text: str) -> None:
        if not self._closed_outbound:
            self._closed_outbound = True
            await self._conn.send_frame(
                _encode_envelope(self.rpc, self.rid, "err", RpcMessage(meta={"error": text}))
            )

    def _feed(self, msg: RpcMessage) -> None:
        self._inbox.put_nowait(msg)

    def _feed_eof(self, exc_text: Optional[str] = None) -> None:
        if not self._finished_inbound:
            self._finished_inbound = True
            self._inbox.put_nowait(exc_text if exc_text is not None else StopAsyncIteration)

    async def receive(self, timeout: Optional[float] = None) -> RpcMessage:
        """Next inbound message; raises RpcError on remote error / EOF."""
        item = await asyncio.wait_for(self._inbox.get(), timeout)
        if item is StopAsyncIteration:
            self._inbox.put_nowait(StopAsyncIteration)  # keep EOF sticky
            raise RpcError(f"stream {self.rpc}#{self.rid}: closed")
        if isinstance(item, str):
            self._inbox.put_nowait(item)
            raise RpcError(f"stream {self.rpc}#{self.rid}: remote error: {item}")
        return item

    def __aiter__(self):
        return self

    async def __anext__(self) -> RpcMessage:
        item = await self._inbox.get()
        if item is StopAsyncIteration:
            self._inbox.put_nowait(StopAsyncIteration)
            raise StopAsyncIteration
        if isinstance(item, str):
            self._inbox.put_nowait(item)
            raise RpcError(f"stream {self.rpc}#{self.rid}: remote error: {item}")
        if item.kind == "end":
            self._finished_inbound = True
            # deliver the final message, next iteration stops
            self._inbox.put_nowait(StopAsyncIteration)
        return item


RpcHandler = Callable[[RpcMessage, RpcStream], Awaitable[None]]

# ---------------------------------------------------------------- in-process
# Registry of P2PNodes listening in THIS process. When a client dials an
# address served by the same process (co-located client+server — the common
# deployment for a GPU node that both serves blocks and runs the thin client),
# the RPC bypasses sockets AND serialization entirely: RpcMessages cross
# loops by reference, so GPU tensors never leave the device. Disable with
# PETALS_AMD_NO_INPROC=1 (used by wire-level tests).

_INPROC_NODES: Dict[Tuple[str, int], "P2PNode"] = {}


def _inproc_enabled() -> bool:
    return not os.environ.get("PETALS_AMD_NO_INPROC")


class InProcStream:
    """One endpoint of an in-process RPC stream. Mirrors RpcStream's interface
    and EOF/error semantics; each endpoint is owned by one asyncio loop and
    fed thread-safely from the peer's loop."""

    _EOF = StopAsyncIteration

    def __init__(self, owner_loop: asyncio.AbstractEventLoop, rpc: str, rid: int):
        self._loop = owner_loop
        self.rpc = rpc
        self.rid = rid
        self._items = __import__("collections").deque()
        self._lock = __import__("threading").Lock()
        self._event: Optional[asyncio.Event] = None  # created lazily ON the owner loop
        self.peer: Optional["InProcStream"] = None
        self._closed_outbound = False
        self.is_inproc = True

    # -- feeding (called from any thread)

    def _deliver(self, item) -> None:
        with self._lock:
            self._items.append(item)
        try:
            self._loop.call_soon_threadsafe(self._notify)
        except RuntimeError:
            pass  # owner loop already closed

    def _notify(self) -> None:
        if self._event is not None:
            self._event.set()

    def _feed_eof(self, exc_text: Optional[str] = None) -> None:
        self._deliver(exc_text if exc_text is not None else self._EOF)

    # -- receiving (called on the owner loop)

    def _pop(self):
        with self._lock:
            return self._items.popleft() if self._items else None

    async def _next_item(self, timeout: Optional[float]):
        if self._event is None:
            self._event = asyncio.Event()
        deadline = None if timeout is None else asyncio.get_event_loop().time() + timeout
        while True:
            self._event.clear()
            item = self._pop()
            if item is not None:
                return item
            remaining = None if deadline is None else deadline - asyncio.get_event_loop().time()
            if remaining is not None and remaining <= 0:
                raise asyncio.TimeoutError()
            await asyncio.wait_for(self._event.wait(), remaining)

    async def receive(self, timeout: Optional[float] = None) -> RpcMessage:
        item = await self._next_item(timeout)
        if item is self._EOF:
            self._deliver(self._EOF)  # sticky
            raise RpcError(f"stream {self.rpc}#{self.rid}: closed")
        if isinstance(item, str):
            self._deliver(item)
            raise RpcError(f"stream {self.rpc}#{self.rid}: remote error: {item}")
        return item

    def __aiter__(self):
        return self

    async def __anext__(self) -> RpcMessage:
        item = await self._next_item(None)
        if item is self._EOF:
            self._deliver(self._EOF)
            raise StopAsyncIteration
        if isinstance(item, str):
            self._deliver(item)
            raise RpcError(f"stream {self.rpc}#{self.rid}: remote error: {item}")
        if item.kind == "end":
            self._deliver(self._EOF)
        return item

    # -- sending

    async def send(self, msg: RpcMessage, kind: str = "item", compressions=None) -> None:
        if self._closed_outbound:
            raise RpcError(f"stream {self.rpc}#{self.rid} already closed for sending")
        if kind == "end":
            self._closed_outbound = True
        if self.peer is not None:
            self.peer._deliver(RpcMessage(meta=dict(msg.meta), tensors=list(msg.tensors), kind=kind))

    async def close(self, msg: Optional[RpcMessage] = None) -> None:
        if not self._closed_outbound:
            await self.send(msg or RpcMessage(), kind="end")

    async def error(self, text: str) -> None:
        if not self._closed_outbound:
            self._closed_outbound = True
            if self.peer is not None:
                self.peer._feed_eof(text)


class Connection:
    """One TCP connection; multiplexes many RPC streams."""

    def __init__(self, node: "P2PNode", reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        self.node = node
        self.reader = reader
        self.writer = writer
        self.streams: Dict[int, RpcStream] = {}
        self._rid_counter = 0
        self._send_lock = asyncio.Lock()
        self._closed = False
        self._reader_task: Optional[asyncio.Task] = None

    def start(self):
        self._reader_task = asyncio.get_event_loop().create_task(self._read_loop())

    @property
    def is_closed(self) -> bool:
        return self._closed

    def next_rid(self) -> int:
        self._rid_counter += 2  # client uses even rids, 0 reserved
        return self._rid_counter

    async def send_frame(self, frame: bytes) -> None:
        if self._closed:
            raise RpcError("connection closed")
        async with self._send_lock:
            self.writer.write(frame)
            await self.writer.drain()

    async def _read_loop(self):
        exc_text = None
        try:
            while True:
                header = await self.reader.readexactly(4)
                (length,) = struct.unpack(">I", header)
                if length > MAX_FRAME_SIZE:
                    raise RpcError(f"frame too large: {length}")
                body = await self.reader.readexactly(length)
                rpc, rid, kind, msg = _decode_envelope(body)
                if kind in ("req", "req_end"):
                    msg.kind = "end" if kind == "req_end" else "item"
                    stream = RpcStream(self, rpc, rid)
                    self.streams[rid] = stream
                    handler = self.node.handlers.get(rpc)
                    if handler is None:
                        await stream.error(f"unknown rpc {rpc!r}")
                        continue
                    asyncio.get_event_loop().create_task(self._run_handler(handler, msg, stream))
                else:
                    stream = self.streams.get(rid)
                    if stream is None:
                        continue  # late message for a finished stream
                    if kind == "err":
                        stream._feed_eof(msg.meta.get("error", "remote error"))
                        self.streams.pop(rid, None)
                    else:
                        stream._feed(msg)
                        if kind == "end":
                            self.streams.pop(rid, None)
        except (asyncio.IncompleteReadError, ConnectionResetError, BrokenPipeError, OSError) as e:
            exc_text = f"connection lost: {type(e).__name__}"
        except Exception as e:  # noqa: BLE001
            exc_text = f"transport error: {e!r}"
            logger.exception("transport read loop failed")
        finally:
            await self._shutdown(exc_text or "connection closed")

    async def _run_handler(self, handler: RpcHandler, first: RpcMessage, stream: RpcStream):
        # NB: the initial request is passed to the handler as an argument and is
        # NOT fed into the stream inbox — the iterator yields only later messages
        try:
            if first.kind == "end":
                stream._feed_eof()
            await handler(first, stream)
            await stream.close()
        except asyncio.CancelledError:
            raise
        except Exception as e:  # noqa: BLE001
            logger.debug("rpc handler %s failed: %r", stream.rpc, e, exc_info=True)
            try:
                await stream.error(f"{type(e).__name__}: {e}")
            except Exception:  # noqa: BLE001
                pass
        finally:
            self.streams.pop(stream.rid, None)

    async def _shutdown(self, reason: str):
        if self._closed:
            return
        self._closed = True
        for stream in list(self.streams.values()):
            stream._feed_eof(reason)
        self.streams.clear()
        try:
            self.writer.close()
        except Exception:  # noqa: BLE001
            pass
        self.node._forget_connection(self)

    async def close(self):
        if self._reader_task is not None:
            self._reader_task.cancel()
        await self._shutdown("closed locally")


MAGIC_PLAIN = b"PAMD1\n"
MAGIC_TLS = b"PAMDS\n"

# process-wide default for P2PNode(secure=None): set from ClientConfig.secure
# or the PETALS_AMD_SECURE env var (a swarm runs either all-secure or all-plain)
DEFAULT_SECURE: Optional[bool] = None


async def _starttls(reader: asyncio.StreamReader, writer: asyncio.StreamWriter,
                    ctx, server_side: bool) -> None:
    """Upgrade an asyncio stream pair to TLS in place (py3.10 recipe: swap the
    transport under the existing StreamReaderProtocol/StreamWriter)."""
    loop = asyncio.get_event_loop()
    transport = writer.transport
    protocol = transport.get_protocol()
    new_transport = await loop.start_tls(transport, protocol, ctx, server_side=server_side)
    writer._transport = new_transport  # noqa: SLF001


class P2PNode:
    """A peer: can listen for inbound RPCs and open outbound RPC streams.

    Replaces the reference's per-process `p2pd` daemon + hivemind.P2P wrapper.
    """

    def __init__(self, peer_id: Optional[str] = None, *, identity=None, secure: Optional[bool] = None):
        if secure is None:
            secure = DEFAULT_SECURE if DEFAULT_SECURE is not None else bool(os.environ.get("PETALS_AMD_SECURE"))
        self.secure = secure
        self.identity = identity
        if secure and self.identity is None:
            from petals_amd.p2p.identity import NodeIdentity

            self.identity = NodeIdentity()
        self.peer_id = (self.identity.peer_id if self.identity is not None else None) or peer_id or os.urandom(16).hex()
        self.handlers: Dict[str, RpcHandler] = {}
        self._server: Optional[asyncio.base_events.Server] = None
        self.listen_addr: Optional[Tuple[str, int]] = None
        self._conns: Dict[tuple, Connection] = {}
        self._inbound: List[Connection] = []
        self._conn_locks: Dict[tuple, asyncio.Lock] = {}
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._inproc_keys: List[Tuple[str, int]] = []
        self._expected_peers: Dict[Tuple[str, int], str] = {}  # addr -> announced peer id
        self._ssl_server_ctx = None
        self._ssl_client_ctx = None

    def add_handler(self, rpc: str, handler: RpcHandler) -> None:
        self.handlers[rpc] = handler

    def expect_peer(self, addr, peer_id: str) -> None:
        """Record the DHT-announced peer id for an address; secure connects
        verify the certificate fingerprint against it."""
        self._expected_peers[(addr[0], int(addr[1]))] = peer_id

    def _server_ctx(self):
        if self._ssl_server_ctx is None and self.identity is not None:
            self._ssl_server_ctx = self.identity.server_ssl_context()
        return self._ssl_server_ctx

    def _client_ctx(self):
        if self._ssl_client_ctx is None:
            from petals_amd.p2p.identity import NodeIdentity

            self._ssl_client_ctx = NodeIdentity.client_ssl_context()
        return self._ssl_client_ctx

    async def listen(self, host: str = "127.0.0.1", port: int = 0) -> Tuple[str, int]:
        self._server = await asyncio.start_server(self._on_inbound, host=host, port=port)
        sockets = self._server.sockets
        addr = sockets[0].getsockname()
        self.listen_addr = (host, addr[1])
        self._loop = asyncio.get_event_loop()
        for h in {host, "127.0.0.1"}:
            key = (h, addr[1])
            _INPROC_NODES[key] = self
            self._inproc_keys.append(key)
        return self.listen_addr

    async def _on_inbound(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        sock = writer.get_extra_info("socket")
        if sock is not None:
            # without NODELAY on the accepted socket, small RPC responses sit in
            # Nagle's buffer against the peer's delayed ACK (~40 ms per step)
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        await self.accept_stream_pair(reader, writer)

    async def accept_stream_pair(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        """Accepting side of the transport handshake — used for direct inbound
        TCP and for relay tunnels (p2p/relay.py) alike."""
        try:
            magic = await asyncio.wait_for(reader.readexactly(len(MAGIC_PLAIN)), 30.0)
            if magic == MAGIC_TLS:
                ctx = self._server_ctx()
                if ctx is None:
                    writer.close()
                    return
                writer.write(b"OK\n")
                await writer.drain()
                await _starttls(reader, writer, ctx, server_side=True)
            elif magic == MAGIC_PLAIN:
                if self.secure:
                    logger.warning("refusing plaintext connection (secure mode)")
                    writer.close()
                    return
            elif magic == b"PAMDR\n":  # relay: client leg asking for a tunnel
                hub = getattr(self, "relay_hub", None)
                target = (await asyncio.wait_for(reader.readline(), 30.0)).decode().strip()
                if hub is None or not target:
                    writer.close()
                    return
                await hub.open_tunnel(target, reader, writer)
                return
            elif magic == b"PAMDT\n":  # relay: target leg dialing back
                hub = getattr(self, "relay_hub", None)
                token = (await asyncio.wait_for(reader.readline(), 30.0)).decode().strip()
                if hub is None or not hub.accept_tunnel(token, reader, writer):
                    writer.close()
                return
            else:
                writer.close()
                return
        except Exception:  # noqa: BLE001
            try:
                writer.close()
            except Exception:  # noqa: BLE001
                pass
            return
        conn = Connection(self, reader, writer)
        self._inbound.append(conn)
        conn.start()

    def _forget_connection(self, conn: Connection):
        for key, val in list(self._conns.items()):
            if val is conn:
                self._conns.pop(key, None)
        if conn in self._inbound:
            self._inbound.remove(conn)

    async def connect(self, addr, timeout: float = 10.0) -> Connection:
        addr = self._norm_addr(addr)
        lock = self._conn_locks.setdefault(addr, asyncio.Lock())
        async with lock:
            conn = self._conns.get(addr)
            if conn is not None and not conn.is_closed:
                return conn
            if len(addr) >= 4 and addr[2] == "relay":
                from petals_amd.p2p.relay import connect_via_relay

                reader, writer = await connect_via_relay(self, (addr[0], addr[1]), addr[3], timeout)
            else:
                reader, writer = await asyncio.wait_for(asyncio.open_connection(addr[0], addr[1]), timeout)
                sock = writer.get_extra_info("socket")
                if sock is not None:
                    sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            conn = await self._handshake_outbound(addr, reader, writer)
            self._conns[addr] = conn
            return conn

    @staticmethod
    def _norm_addr(addr) -> tuple:
        if len(addr) >= 4 and addr[2] == "relay":
            return (addr[0], int(addr[1]), "relay", addr[3])
        return (addr[0], int(addr[1]))

    async def _handshake_outbound(self, addr, reader, writer) -> Connection:
        """Dialer side of the magic/STARTTLS exchange + peer verification."""
        expected = None
        if len(addr) >= 4 and addr[2] == "relay":
            expected = addr[3]  # the relay target IS the peer we must reach
        else:
            expected = self._expected_peers.get((addr[0], addr[1]))
        if self.secure:
            writer.write(MAGIC_TLS)
            await writer.drain()
            ok = await asyncio.wait_for(reader.readexactly(3), 30.0)
            if ok != b"OK\n":
                writer.close()
                raise RpcError(f"peer at {addr} refused STARTTLS")
            await _starttls(reader, writer, self._client_ctx(), server_side=False)
            ssl_obj = writer.transport.get_extra_info("ssl_object")
            der = ssl_obj.getpeercert(binary_form=True) if ssl_obj else None
            if der is None:
                writer.close()
                raise RpcError("TLS peer presented no certificate")
            from petals_amd.p2p.identity import cert_fingerprint

            actual = cert_fingerprint(der)
            if expected is not None and actual != expected:
                writer.close()
                raise RpcError(
                    f"peer identity mismatch at {addr}: announced {expected[:8]}, "
                    f"certificate fingerprint {actual[:8]} — possible impersonation"
                )
        else:
            writer.write(MAGIC_PLAIN)
            await writer.drain()
        conn = Connection(self, reader, writer)
        conn.start()
        return conn

    async def open_stream(
        self,
        addr: Tuple[str, int],
        rpc: str,
        request: RpcMessage,
        timeout: float = 10.0,
        compressions=None,
        end: bool = False,
    ) -> RpcStream:
        """Send the initial request message and return the live stream.

        With ``end=True`` the request also closes our outbound side (unary
        request)."""
        addr = self._norm_addr(addr)
        target = _INPROC_NODES.get(addr) if _inproc_enabled() and len(addr) == 2 else None
        if target is not None and target._loop is not None and not target._loop.is_closed():
            return self._open_inproc(target, rpc, request, end)
        conn = await self.connect(addr, timeout)
        rid = conn.next_rid()
        stream = RpcStream(conn, rpc, rid)
        conn.streams[rid] = stream
        if end:
            stream._closed_outbound = True
        await conn.send_frame(_encode_envelope(rpc, rid, "req_end" if end else "req", request, compressions))
        return stream

    def _open_inproc(self, target: "P2PNode", rpc: str, request: RpcMessage, end: bool) -> InProcStream:
        local = InProcStream(asyncio.get_event_loop(), rpc, id(request))
        remote = InProcStream(target._loop, rpc, id(request))
        local.peer = remote
        remote.peer = local
        if end:
            local._closed_outbound = True
        first = RpcMessage(
            meta=dict(request.meta), tensors=list(request.tensors), kind="end" if end else "item"
        )

        def _start():
            handler = target.handlers.get(rpc)
            if handler is None:
                local._feed_eof(f"unknown rpc {rpc!r}")
                return
            if first.kind == "end":
                remote._feed_eof()
            asyncio.ensure_future(_run_inproc_handler(handler, first, remote))

        target._loop.call_soon_threadsafe(_start)
        return local

    async def call_unary(
        self, addr: Tuple[str, int], rpc: str, request: RpcMessage, timeout: float = 30.0, compressions=None
    ) -> RpcMessage:
        stream = await self.open_stream(
            addr, rpc, request, timeout=timeout, compressions=compressions, end=True
        )
        return await stream.receive(timeout=timeout)

    async def shutdown(self):
        for key in self._inproc_keys:
            if _INPROC_NODES.get(key) is self:
                _INPROC_NODES.pop(key, None)
        self._inproc_keys.clear()
        for conn in list(self._conns.values()) + list(self._inbound):
            await conn.close()
        if self._server is not None:
            self._server.close()
            try:
                await self._server.wait_closed()
            except Exception:  # noqa: BLE001
                pass


async def _run_inproc_handler(handler: RpcHandler, first: RpcMessage, stream: InProcStream):
    try:
        await handler(first, stream)
        await stream.close()
    except asyncio.CancelledError:
        raise
    except Exception as e:  # noqa: BLE001
        logger.debug("in-proc rpc handler %s failed: %r", stream.rpc, e, exc_info=True)
        try:
            await stream.error(f"{type(e).__name__}: {e}")
        except Exception:  # noqa: BLE001
            pass
