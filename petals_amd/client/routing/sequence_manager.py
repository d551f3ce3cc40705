"""Client-side routing brain.

Parity with reference ``client/routing/sequence_manager.py`` (:59-543):
background DHT refresh, server blacklisting with exponential backoff, RTT
pings, and two route modes — "min_latency" (Dijkstra over (block) nodes with
RTT + compute + cache-allocation edge costs) and "max_throughput"
(throughput-weighted random spans).
"""

from __future__ import annotations

import dataclasses
import logging
import math
import random
import threading
import time
from typing import Any, Dict, List, Optional, Sequence, Tuple

from petals_amd.client.config import ClientConfig
from petals_amd.client.routing.sequence_info import RemoteSequenceInfo
from petals_amd.data_structures import ModuleUID, RemoteSpanInfo
from petals_amd.dht.node import DHT
from petals_amd.utils.dht import get_remote_module_infos
from petals_amd.utils.ping import PingAggregator

logger = logging.getLogger(__name__)


class MissingBlocksError(RuntimeError):
    def __init__(self, blocks):
        super().__init__(f"no servers holding blocks {blocks}; swarm is incomplete or still joining")


@dataclasses.dataclass
class _Ban:
    banned_until: float
    n_fails: int


class RemoteSequenceManager:
    """Keeps the swarm view fresh and builds server chains for requests."""

    def __init__(
        self,
        config: ClientConfig,
        block_uids: Sequence[ModuleUID],
        *,
        dht: Optional[DHT] = None,
    ):
        self.config = config
        if config.secure is not None:
            from petals_amd.p2p import transport as _transport

            _transport.DEFAULT_SECURE = config.secure
        self.block_uids = tuple(block_uids)
        self.dht = dht or DHT(initial_peers=config.initial_peers, client_mode=True)
        self._owns_dht = dht is None
        self.state = RemoteSequenceInfo.make_empty(self.block_uids)
        self.addrs: Dict[str, Tuple[str, int]] = {}
        self._lock = threading.Lock()
        self._bans: Dict[str, _Ban] = {}
        self.ping_aggregator = PingAggregator(self.dht.node.p2p)
        self._last_update = 0.0
        self._update_thread: Optional[threading.Thread] = None
        self._stop = threading.Event()
        self._session_id_counter = random.Random()

    # ------------------------------------------------------------- updates

    def _filter_servers(self, infos) -> None:
        """Drop banned / not-allowed / blocked servers in place (client policy:
        reference sequence_manager.py bans + ClientConfig allow/block lists)."""
        now = time.monotonic()
        for info in infos:
            if info is None:
                continue
            for peer_id in list(info.servers.keys()):
                ban = self._bans.get(peer_id)
                if ban is not None and ban.banned_until > now:
                    del info.servers[peer_id]
                elif self.config.allowed_servers is not None and peer_id not in self.config.allowed_servers:
                    del info.servers[peer_id]
                elif self.config.blocked_servers is not None and peer_id in self.config.blocked_servers:
                    del info.servers[peer_id]

    def update(self, wait: bool = True) -> None:
        infos, addrs = get_remote_module_infos(self.dht, self.block_uids)
        with self._lock:
            self._filter_servers(infos)
            self.state.update_([i if (i and i.servers) else None for i in infos])
            self.addrs.update(addrs)
            self._last_update = time.monotonic()
        try:
            # secure transports verify certificate fingerprints against the
            # DHT-announced peer ids at connect time
            p2p = self.p2p
            for pid, addr in addrs.items():
                if len(addr) == 2:
                    p2p.expect_peer(addr, pid)
        except Exception:  # noqa: BLE001
            pass

    def _maybe_update(self):
        if time.monotonic() - self._last_update > self.config.update_period or not self.state.spans_by_priority:
            self.update()

    def start_background_updates(self):
        if self._update_thread is not None:
            return

        def loop():
            while not self._stop.wait(self.config.update_period):
                try:
                    self.update()
                except Exception as e:  # noqa: BLE001
                    logger.warning("background update failed: %r", e)

        self._update_thread = threading.Thread(target=loop, name="SeqManagerUpdate", daemon=True)
        self._update_thread.start()

    # ------------------------------------------------------------- routing

    def make_sequence(
        self,
        start_index: int = 0,
        end_index: Optional[int] = None,
        *,
        mode: str = "max_throughput",
        cache_tokens_needed: Optional[int] = None,
    ) -> List[RemoteSpanInfo]:
        end_index = end_index if end_index is not None else len(self.block_uids)
        self._maybe_update()
        for attempt in range(2):
            try:
                if mode == "min_latency":
                    seq = self._make_sequence_min_latency(start_index, end_index, cache_tokens_needed)
                elif mode == "max_throughput":
                    seq = self._make_sequence_max_throughput(start_index, end_index)
                else:
                    raise ValueError(f"unknown route mode {mode!r}")
                if self.config.show_route is True or (self.config.show_route == "inference" and mode == "min_latency"):
                    route = " => ".join(f"{s.peer_id[:8]}[{s.start}:{s.end}]" for s in seq)
                    logger.info("route found: %s", route)
                return seq
            except MissingBlocksError:
                if attempt == 0:
                    self.update()
                else:
                    raise
        raise MissingBlocksError(list(range(start_index, end_index)))

    def _spans_at(self, block_idx: int) -> List[RemoteSpanInfo]:
        return self.state.spans_containing_block[block_idx]

    def _make_sequence_max_throughput(self, start: int, end: int) -> List[RemoteSpanInfo]:
        """Random server per hop, weighted by throughput (parity :302-324)."""
        client_server_rtt = 0.0  # placeholder; throughput dominates
        span_sequence: List[RemoteSpanInfo] = []
        current = start
        while current < end:
            candidates = [s for s in self._spans_at(current) if s.start <= current < s.end]
            if not candidates:
                raise MissingBlocksError([current])
            weights = [max(s.throughput, 1e-9) for s in candidates]
            chosen = random.choices(candidates, weights=weights, k=1)[0]
            chosen = dataclasses.replace(chosen)  # do not mutate the shared state
            chosen.start = current
            span_sequence.append(chosen)
            current = min(chosen.end, end)
        return span_sequence

    @staticmethod
    def _rtt_to_delay(rtt: Optional[float], *, default_delay: float = 0.15, max_delay: float = 5.0) -> float:
        if rtt is None:
            return default_delay
        return min(rtt / 2, max_delay)

    @staticmethod
    def _has_cache_for(span: RemoteSpanInfo, cache_tokens_needed: Optional[int]) -> bool:
        if cache_tokens_needed is None or span.server_info.cache_tokens_left is None:
            return True
        # pessimistic: assume the whole hosted span is used (false positives
        # cost more than false negatives here)
        return cache_tokens_needed * 2 * span.length <= span.server_info.cache_tokens_left

    def _make_sequence_min_latency(
        self,
        start: int,
        end: int,
        cache_tokens_needed: Optional[int],
        *,
        overhead_delay: float = 0.018,  # serialization overhead per hop
        default_inference_rps: float = 300.0,
        alloc_delay: float = 10.0,  # penalty when the server must evict cache first
    ) -> List[RemoteSpanInfo]:
        """Dijkstra over (peer, block) nodes, mirroring the reference cost
        model (reference sequence_manager.py:177-289): client<->server RTTs
        price the first and last hops; chained server->server hops are priced
        with the UPSTREAM server's gossiped `next_pings` RTT to the downstream
        server (announced by ModuleAnnouncer pings, reference server.py:760);
        compute edges cost 1/inference_rps per block; a chosen server is used
        to its span end (keeps the graph O(servers * blocks))."""
        import heapq

        missing = [i for i in range(start, end) if not self._spans_at(i)]
        if missing:
            raise MissingBlocksError(missing)

        client_rtts = self.ping_aggregator.to_dict()
        spans_at: Dict[int, List[RemoteSpanInfo]] = {i: self._spans_at(i) for i in range(start, end)}
        span_of: Dict[str, RemoteSpanInfo] = {}
        for i in range(start, end):
            for s in spans_at[i]:
                span_of[s.peer_id] = s

        # adjacency: node -> list of (cost, node). Nodes: "start", "end",
        # (peer_id, block_idx)
        adj: Dict[Any, List[Tuple[float, Any]]] = {}

        def add_edge(u, v, c):
            adj.setdefault(u, []).append((c, v))

        for span in spans_at[start]:
            delay = self._rtt_to_delay(client_rtts.get(span.peer_id)) + overhead_delay
            if not self._has_cache_for(span, cache_tokens_needed):
                delay += alloc_delay
            add_edge("start", (span.peer_id, start), delay)
        for span in spans_at[end - 1]:
            add_edge((span.peer_id, end), "end", self._rtt_to_delay(client_rtts.get(span.peer_id)))
        for block_idx in range(start + 1, end):
            for cur in spans_at[block_idx - 1]:
                if cur.end != block_idx:
                    continue  # a chosen server is used to its span end
                for nxt in spans_at[block_idx]:
                    rtt = None
                    if cur.server_info.next_pings is not None:
                        rtt = cur.server_info.next_pings.get(nxt.peer_id)
                    delay = self._rtt_to_delay(rtt) + overhead_delay
                    if not self._has_cache_for(nxt, cache_tokens_needed):
                        delay += alloc_delay
                    add_edge((cur.peer_id, block_idx), (nxt.peer_id, block_idx), delay)
        for span in span_of.values():
            rps = span.server_info.inference_rps or span.throughput or default_inference_rps
            for block_idx in range(max(span.start, start), min(span.end, end)):
                add_edge((span.peer_id, block_idx), (span.peer_id, block_idx + 1), 1.0 / max(rps, 1e-9))

        dist: Dict[Any, float] = {"start": 0.0}
        prev: Dict[Any, Any] = {}
        heap = [(0.0, 0, "start")]
        tie = 0
        done = set()
        while heap:
            d, _, u = heapq.heappop(heap)
            if u in done:
                continue
            done.add(u)
            if u == "end":
                break
            for c, v in adj.get(u, ()):
                nd = d + c
                if nd < dist.get(v, math.inf):
                    dist[v] = nd
                    prev[v] = u
                    tie += 1
                    heapq.heappush(heap, (nd, tie, v))
        if "end" not in done:
            raise MissingBlocksError(list(range(start, end)))

        # walk back: nodes between "start" and "end" are (peer, block)
        path = []
        node = prev["end"]
        while node != "start":
            path.append(node)
            node = prev[node]
        path.reverse()
        seq: List[RemoteSpanInfo] = []
        for peer_id, block_idx in path:
            if not seq or seq[-1].peer_id != peer_id:
                span = dataclasses.replace(span_of[peer_id])
                span.start = span.end = block_idx
                seq.append(span)
            else:
                seq[-1].end = block_idx
        # drop empty spans (measurement-noise triangle-inequality violations)
        return [s for s in seq if s.length > 0]

    # ------------------------------------------------------------ failures

    def on_request_failure(self, peer_id: Optional[str]):
        if peer_id is None:
            return
        with self._lock:
            ban = self._bans.get(peer_id)
            n_fails = (ban.n_fails + 1) if ban else 1
            duration = min(self.config.ban_timeout * (2 ** (n_fails - 1)), 300.0)
            self._bans[peer_id] = _Ban(time.monotonic() + duration, n_fails)
        logger.debug("banned %s for %.1f s", peer_id[:8], duration)
        self.update()

    def on_request_success(self, peer_id: str):
        with self._lock:
            self._bans.pop(peer_id, None)

    def get_retry_delay(self, attempt_no: int) -> float:
        if attempt_no == 0:
            return 0.0
        return min(self.config.min_backoff * 2 ** (attempt_no - 1), self.config.max_backoff)

    def get_request_metadata(self, protocol: str, *args, **kwargs) -> Dict[str, Any]:
        meta = {
            "active_adapter": self.config.active_adapter,
            "points": 0,
        }
        if self.config.output_compression != "none":
            meta["output_compression"] = self.config.output_compression
        return meta

    def address_of(self, peer_id: str) -> Tuple[str, int]:
        addr = self.addrs.get(peer_id)
        if addr is None:
            raise KeyError(f"no known address for peer {peer_id[:8]}")
        return addr

    def ping_sequence(self, spans: Sequence[RemoteSpanInfo]):
        """Refresh RTTs for candidate servers (best effort)."""
        peers = {s.peer_id: self.addrs[s.peer_id] for s in spans if s.peer_id in self.addrs}
        try:
            self.dht.run_coroutine(self.ping_aggregator.ping(peers), timeout=10)
        except Exception:  # noqa: BLE001
            pass

    @property
    def num_blocks(self) -> int:
        return len(self.block_uids)

    def run_coroutine(self, coro, timeout: Optional[float] = None):
        """Client RPC coroutines run on the RemoteWorker loop (NOT the DHT's —
        routing calls inside them block on DHT lookups; a shared loop would
        deadlock)."""
        from petals_amd.client.remote_worker import get_worker

        return get_worker().run_coroutine(coro, timeout)

    @property
    def p2p(self):
        from petals_amd.client.remote_worker import get_worker

        worker = get_worker()
        worker._ensure_started()
        return worker.p2p

    def shutdown(self):
        self._stop.set()
        if self._update_thread is not None:
            self._update_thread.join(timeout=5)
        if self._owns_dht:
            self.dht.shutdown()
