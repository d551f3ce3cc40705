"""Kademlia-style DHT over the petals_amd TCP transport.

Replaces the reference's hivemind DHT (Python atop the go-libp2p daemon; see
SURVEY §2.3). Same key schema as the reference so block discovery semantics
carry over: key = module uid "{prefix}.{i}", subkey = peer_id, value =
ServerInfo dict, per-subkey expiration (reference `utils/dht.py:28-71`).

Design:
  * 128-bit node ids (the transport peer id), XOR metric, k-buckets (k=8)
  * unary RPCs: dht.ping / dht.find_node / dht.store / dht.find_value
  * iterative lookups with alpha=3 parallelism
  * store_many() writes to the K closest nodes (and locally);
    get_many() merges {subkey: (value, expiration)} maps across the K closest
  * all methods are asyncio coroutines; the `DHT` wrapper runs a private event
    loop thread so synchronous server/client code can call in

A node in *client mode* still listens on an ephemeral local port (needed for
response routing) but never announces itself into other nodes' routing tables
as a storage target.
"""

from __future__ import annotations

import asyncio
import heapq
import logging
import threading
import time
from typing import Any, Dict, List, Optional, Sequence, Tuple

from petals_amd.p2p.transport import P2PNode, RpcError, RpcMessage

logger = logging.getLogger(__name__)

K_BUCKET = 8
ALPHA = 3
ID_BITS = 128

Addr = Tuple[str, int]
PeerTriple = Tuple[str, str, int]  # (peer_id hex, host, port)


def _distance(a: str, b: str) -> int:
    return int(a, 16) ^ int(b, 16)


class RoutingTable:
    def __init__(self, own_id: str):
        self.own_id = own_id
        self.buckets: List[List[PeerTriple]] = [[] for _ in range(ID_BITS)]
        self.addr_of: Dict[str, Addr] = {}

    def _bucket_index(self, peer_id: str) -> int:
        d = _distance(self.own_id, peer_id)
        return max(0, d.bit_length() - 1)

    def add(self, peer_id: str, host: str, port: int) -> None:
        if peer_id == self.own_id:
            return
        self.addr_of[peer_id] = (host, port)
        bucket = self.buckets[self._bucket_index(peer_id)]
        for i, (pid, _, _) in enumerate(bucket):
            if pid == peer_id:
                bucket[i] = (peer_id, host, port)
                return
        if len(bucket) < K_BUCKET:
            bucket.append((peer_id, host, port))
        else:
            bucket.pop(0)  # LRU-ish: evict oldest (no ping-before-evict in v1)
            bucket.append((peer_id, host, port))

    def remove(self, peer_id: str) -> None:
        bucket = self.buckets[self._bucket_index(peer_id)]
        self.buckets[self._bucket_index(peer_id)] = [t for t in bucket if t[0] != peer_id]
        self.addr_of.pop(peer_id, None)

    def nearest(self, target_id: str, k: int = K_BUCKET) -> List[PeerTriple]:
        all_peers = [t for bucket in self.buckets for t in bucket]
        return heapq.nsmallest(k, all_peers, key=lambda t: _distance(t[0], target_id))


def _key_id(key: str) -> str:
    import hashlib

    return hashlib.blake2b(key.encode(), digest_size=16).hexdigest()


class DHTNode:
    """Asyncio Kademlia node. Construct via `await DHTNode.create(...)`."""

    def __init__(self, p2p: P2PNode, client_mode: bool = False):
        self.p2p = p2p
        self.client_mode = client_mode
        self.table = RoutingTable(p2p.peer_id)
        # key -> subkey -> (value, expiration_time)
        self.storage: Dict[str, Dict[Any, Tuple[Any, float]]] = {}
        self._cleanup_task: Optional[asyncio.Task] = None
        for rpc, fn in (
            ("dht.ping", self._rpc_ping),
            ("dht.find_node", self._rpc_find_node),
            ("dht.store", self._rpc_store),
            ("dht.find_value", self._rpc_find_value),
        ):
            p2p.add_handler(rpc, fn)

    @classmethod
    async def create(
        cls,
        initial_peers: Sequence[Addr] = (),
        host: str = "127.0.0.1",
        port: int = 0,
        client_mode: bool = False,
        p2p: Optional[P2PNode] = None,
    ) -> "DHTNode":
        p2p = p2p or P2PNode()
        node = cls(p2p, client_mode=client_mode)
        if p2p.listen_addr is None:
            await p2p.listen(host=host, port=port)
        for addr in initial_peers:
            try:
                resp = await node._call(tuple(addr), "dht.ping", {})
                node.table.add(resp["peer_id"], addr[0], addr[1])
            except (RpcError, OSError, asyncio.TimeoutError) as e:
                logger.warning("bootstrap peer %s unreachable: %r", addr, e)
        if node.table.addr_of:
            await node.lookup_nodes(p2p.peer_id)  # populate routing table
        node._cleanup_task = asyncio.get_event_loop().create_task(node._cleanup_loop())
        return node

    @property
    def peer_id(self) -> str:
        return self.p2p.peer_id

    @property
    def listen_addr(self) -> Addr:
        assert self.p2p.listen_addr is not None
        return self.p2p.listen_addr

    # ------------------------------------------------------------------ RPCs

    def _own_triple(self) -> Optional[PeerTriple]:
        if self.client_mode or self.p2p.listen_addr is None:
            return None
        return (self.p2p.peer_id, *self.p2p.listen_addr)

    async def _rpc_ping(self, request: RpcMessage, stream) -> None:
        self._maybe_add_sender(request)
        await stream.close(RpcMessage(meta={"peer_id": self.p2p.peer_id}))

    async def _rpc_find_node(self, request: RpcMessage, stream) -> None:
        self._maybe_add_sender(request)
        target = request.meta["target"]
        peers = self.table.nearest(target, K_BUCKET)
        await stream.close(RpcMessage(meta={"peer_id": self.p2p.peer_id, "peers": [list(t) for t in peers]}))

    async def _rpc_store(self, request: RpcMessage, stream) -> None:
        self._maybe_add_sender(request)
        now = time.time()
        n_stored = 0
        for key, subkey, value, expiration in request.meta["entries"]:
            if expiration <= now:
                continue
            slot = self.storage.setdefault(key, {})
            prev = slot.get(subkey)
            if prev is None or prev[1] <= expiration:
                slot[subkey] = (value, expiration)
                n_stored += 1
        await stream.close(RpcMessage(meta={"peer_id": self.p2p.peer_id, "stored": n_stored}))

    async def _rpc_find_value(self, request: RpcMessage, stream) -> None:
        self._maybe_add_sender(request)
        key = request.meta["key"]
        now = time.time()
        slot = self.storage.get(key, {})
        live = {sk: [v, exp] for sk, (v, exp) in slot.items() if exp > now}
        peers = self.table.nearest(_key_id(key), K_BUCKET)
        await stream.close(
            RpcMessage(
                meta={
                    "peer_id": self.p2p.peer_id,
                    "value": live,
                    "peers": [list(t) for t in peers],
                }
            )
        )

    def _maybe_add_sender(self, request: RpcMessage) -> None:
        sender = request.meta.get("sender")
        if sender:
            self.table.add(sender[0], sender[1], sender[2])

    async def _call(self, addr: Addr, rpc: str, meta: Dict[str, Any], timeout: float = 5.0) -> Dict[str, Any]:
        own = self._own_triple()
        if own is not None:
            meta = {**meta, "sender": list(own)}
        resp = await self.p2p.call_unary(tuple(addr), rpc, RpcMessage(meta=meta), timeout=timeout)
        return resp.meta

    # ------------------------------------------------------------- lookups

    async def lookup_nodes(self, target_id: str, k: int = K_BUCKET) -> List[PeerTriple]:
        """Iterative FIND_NODE: returns up to k closest live peers to target."""
        candidates: Dict[str, PeerTriple] = {t[0]: t for t in self.table.nearest(target_id, k * 2)}
        queried: set = set()
        failed: set = set()
        while True:
            frontier = [
                t
                for pid, t in sorted(candidates.items(), key=lambda kv: _distance(kv[0], target_id))
                if pid not in queried and pid not in failed
            ][:ALPHA]
            if not frontier:
                break
            results = await asyncio.gather(
                *(self._call((t[1], t[2]), "dht.find_node", {"target": target_id}) for t in frontier),
                return_exceptions=True,
            )
            for t, res in zip(frontier, results):
                queried.add(t[0])
                if isinstance(res, BaseException):
                    failed.add(t[0])
                    self.table.remove(t[0])
                    continue
                self.table.add(t[0], t[1], t[2])
                for pid, host, port in (tuple(p) for p in res.get("peers", [])):
                    if pid != self.p2p.peer_id and pid not in candidates:
                        candidates[pid] = (pid, host, port)
        live = [t for pid, t in candidates.items() if pid in queried and pid not in failed]
        live.sort(key=lambda t: _distance(t[0], target_id))
        return live[:k]

    # ---------------------------------------------------------- store / get

    async def store_many(
        self, entries: Sequence[Tuple[str, Any, Any, float]], timeout: float = 5.0
    ) -> int:
        """entries: (key, subkey, value, expiration_time). Returns #peers written."""
        now = time.time()
        # always keep a local replica (a server must see its own announcements)
        by_key: Dict[str, List[Tuple[str, Any, Any, float]]] = {}
        for key, subkey, value, expiration in entries:
            slot = self.storage.setdefault(key, {})
            prev = slot.get(subkey)
            if prev is None or prev[1] <= expiration:
                slot[subkey] = (value, expiration)
            by_key.setdefault(key, []).append((key, subkey, value, expiration))

        n_written = 0
        targets_cache: Dict[str, List[PeerTriple]] = {}
        sends: Dict[Addr, List[Tuple[str, Any, Any, float]]] = {}
        for key, key_entries in by_key.items():
            kid = _key_id(key)
            if kid not in targets_cache:
                targets_cache[kid] = await self.lookup_nodes(kid)
            for pid, host, port in targets_cache[kid]:
                sends.setdefault((host, port), []).extend(key_entries)
        results = await asyncio.gather(
            *(
                self._call(addr, "dht.store", {"entries": [list(e) for e in key_entries]}, timeout)
                for addr, key_entries in sends.items()
            ),
            return_exceptions=True,
        )
        for res in results:
            if not isinstance(res, BaseException):
                n_written += 1
        return n_written

    async def get_many(
        self, keys: Sequence[str], timeout: float = 5.0
    ) -> Dict[str, Dict[Any, Tuple[Any, float]]]:
        """Returns {key: {subkey: (value, expiration)}} merged across replicas."""
        out: Dict[str, Dict[Any, Tuple[Any, float]]] = {}

        async def fetch(key: str):
            merged: Dict[Any, Tuple[Any, float]] = {}
            now = time.time()
            for sk, (v, exp) in self.storage.get(key, {}).items():
                if exp > now:
                    merged[sk] = (v, exp)
            peers = await self.lookup_nodes(_key_id(key))
            results = await asyncio.gather(
                *(self._call((h, p), "dht.find_value", {"key": key}, timeout) for _, h, p in peers),
                return_exceptions=True,
            )
            for res in results:
                if isinstance(res, BaseException):
                    continue
                for sk, (v, exp) in (res.get("value") or {}).items():
                    if exp > now and (sk not in merged or merged[sk][1] < exp):
                        merged[sk] = (v, exp)
            out[key] = merged

        await asyncio.gather(*(fetch(k) for k in keys))
        return out

    async def _cleanup_loop(self):
        while True:
            await asyncio.sleep(30.0)
            now = time.time()
            for key in list(self.storage.keys()):
                slot = self.storage[key]
                for sk in list(slot.keys()):
                    if slot[sk][1] <= now:
                        del slot[sk]
                if not slot:
                    del self.storage[key]

    async def shutdown(self):
        if self._cleanup_task is not None:
            self._cleanup_task.cancel()
        await self.p2p.shutdown()


class DHT:
    """Thread-wrapped DHT node: runs a private asyncio loop so synchronous
    server/client code can use it (parity with `hivemind.DHT(start=True)`).

    Also exposes `run_coroutine()` so other subsystems (RPC clients) can share
    the same event loop thread.
    """

    def __init__(
        self,
        initial_peers: Sequence[Addr] = (),
        host: str = "127.0.0.1",
        port: int = 0,
        client_mode: bool = False,
        start: bool = True,
    ):
        self._initial_peers = [tuple(p) for p in initial_peers]
        self._host, self._port = host, port
        self._client_mode = client_mode
        self.node: Optional[DHTNode] = None
        self.loop: Optional[asyncio.AbstractEventLoop] = None
        self._thread: Optional[threading.Thread] = None
        self._ready = threading.Event()
        if start:
            self.start()

    def start(self):
        self._thread = threading.Thread(target=self._run, name="DHT", daemon=True)
        self._thread.start()
        self._ready.wait(timeout=30)
        if self.node is None:
            raise RuntimeError("DHT failed to start")

    def _run(self):
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
        self.loop = loop

        async def boot():
            self.node = await DHTNode.create(
                initial_peers=self._initial_peers,
                host=self._host,
                port=self._port,
                client_mode=self._client_mode,
            )
            if not self._client_mode:
                # bootstrap/visible DHT nodes double as circuit relays for
                # NAT'd servers (p2p/relay.py)
                from petals_amd.p2p.relay import RelayHub

                RelayHub(self.node.p2p)

        loop.run_until_complete(boot())
        self._ready.set()
        loop.run_forever()
        # drain pending tasks on shutdown
        pending = asyncio.all_tasks(loop)
        for task in pending:
            task.cancel()
        try:
            loop.run_until_complete(asyncio.gather(*pending, return_exceptions=True))
        finally:
            loop.close()

    def run_coroutine(self, coro, timeout: Optional[float] = None):
        assert self.loop is not None
        future = asyncio.run_coroutine_threadsafe(coro, self.loop)
        return future.result(timeout)

    @property
    def peer_id(self) -> str:
        assert self.node is not None
        return self.node.peer_id

    @property
    def listen_addr(self) -> Addr:
        assert self.node is not None
        return self.node.listen_addr

    def store_many(self, entries, timeout: float = 5.0) -> int:
        return self.run_coroutine(self.node.store_many(entries, timeout), timeout=timeout + 10)

    def get_many(self, keys, timeout: float = 5.0):
        return self.run_coroutine(self.node.get_many(keys, timeout), timeout=timeout + 10)

    def shutdown(self):
        if self.loop is None:
            return
        try:
            self.run_coroutine(self.node.shutdown(), timeout=10)
        except Exception:  # noqa: BLE001
            pass
        self.loop.call_soon_threadsafe(self.loop.stop)
        if self._thread is not None:
            self._thread.join(timeout=10)
