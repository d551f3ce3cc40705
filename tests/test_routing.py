"""Routing unit tests (reference tests/test_sequence_manager.py + block
selection semantics), without a live swarm: the manager state is populated
directly from synthetic RemoteModuleInfo."""

import dataclasses
import time

import pytest
import torch

from petals_amd.client.config import ClientConfig
from petals_amd.client.routing.sequence_info import RemoteSequenceInfo
from petals_amd.client.routing.sequence_manager import MissingBlocksError, RemoteSequenceManager
from petals_amd.data_structures import RemoteModuleInfo, ServerInfo, ServerState
from petals_amd.server import block_selection


def make_infos(n_blocks, spans):
    """spans: {peer_id: (start, end, throughput)}"""
    infos = []
    for i in range(n_blocks):
        servers = {}
        for pid, (s, e, thr) in spans.items():
            if s <= i < e:
                servers[pid] = ServerInfo(state=ServerState.ONLINE, throughput=thr, inference_rps=thr,
                                          start_block=s, end_block=e)
        infos.append(RemoteModuleInfo(uid=f"m.{i}", servers=servers) if servers else None)
    return infos


class _FakeManager(RemoteSequenceManager):
    """Bypass the DHT: state is injected."""

    def __init__(self, n_blocks, spans, **cfg):
        self.config = ClientConfig(**cfg)
        self.block_uids = tuple(f"m.{i}" for i in range(n_blocks))
        self.state = RemoteSequenceInfo.make_empty(self.block_uids)
        self.addrs = {pid: ("127.0.0.1", 1) for pid in spans}
        self._bans = {}
        self._lock = __import__("threading").Lock()
        self._last_update = time.monotonic()
        self._update_thread = None
        self.ping_aggregator = type("P", (), {"to_dict": staticmethod(lambda: {})})()
        self.state.update_(make_infos(n_blocks, spans))
        self._spans_src = spans

    def update(self, wait=True):
        infos = make_infos(len(self.block_uids), self._spans_src)
        self._filter_servers(infos)  # the REAL ban/allow/block policy
        self.state.update_([i if (i and i.servers) else None for i in infos])
        self._last_update = time.monotonic()


def test_max_throughput_covers_all_blocks():
    mgr = _FakeManager(8, {"A": (0, 4, 1.0), "B": (4, 8, 1.0), "C": (2, 6, 1.0)})
    for _ in range(5):
        seq = mgr.make_sequence(mode="max_throughput")
        covered = []
        for span in seq:
            assert span.start == (covered[-1] if covered else 0)
            covered.append(span.end)
        assert covered[-1] == 8


def test_min_latency_prefers_fast_full_span():
    mgr = _FakeManager(8, {"slow1": (0, 4, 1.0), "slow2": (4, 8, 1.0), "fast": (0, 8, 100.0)})
    seq = mgr.make_sequence(mode="min_latency")
    assert [s.peer_id for s in seq] == ["fast"]


def test_min_latency_uses_server_to_server_next_pings():
    """Chained hops are priced with the UPSTREAM server's gossiped next_pings
    RTT, not the client's RTT: with asymmetric server->server latencies the
    router must pick the chain whose HOP is fast, even when the client's RTT
    to both candidates is identical."""
    mgr = _FakeManager(
        8,
        {
            "head": (0, 4, 50.0),
            "near": (4, 8, 50.0),  # fast hop from head
            "far": (4, 8, 50.0),   # slow hop from head (same compute/throughput)
        },
    )
    # inject next_pings on the head server's gossip: near is 1 ms away, far 2 s
    for info in mgr.state.block_infos:
        if info and "head" in info.servers:
            info.servers["head"].next_pings = {"near": 0.001, "far": 2.0}
    mgr.state.update_(mgr.state.block_infos)
    seq = mgr.make_sequence(mode="min_latency")
    assert [s.peer_id for s in seq] == ["head", "near"], seq

    # flip the RTTs: the router must follow
    for info in mgr.state.block_infos:
        if info and "head" in info.servers:
            info.servers["head"].next_pings = {"near": 2.0, "far": 0.001}
    mgr.state.update_(mgr.state.block_infos)
    seq = mgr.make_sequence(mode="min_latency")
    assert [s.peer_id for s in seq] == ["head", "far"], seq


def test_missing_blocks_raises():
    mgr = _FakeManager(8, {"A": (0, 4, 1.0)})
    with pytest.raises(MissingBlocksError):
        mgr.make_sequence(mode="max_throughput")


def test_ban_then_recover():
    mgr = _FakeManager(4, {"A": (0, 4, 1.0), "B": (0, 4, 1.0)}, ban_timeout=0.2)
    mgr.on_request_failure("A")
    for _ in range(5):
        seq = mgr.make_sequence(mode="max_throughput")
        assert all(s.peer_id == "B" for s in seq)
    time.sleep(0.25)
    mgr.update()
    peers = {mgr.make_sequence(mode="max_throughput")[0].peer_id for _ in range(20)}
    assert "A" in peers  # ban expired


def test_choose_best_blocks_picks_least_covered():
    infos = make_infos(8, {"A": (0, 4, 1.0)})
    chosen = block_selection.choose_best_blocks(4, infos)
    assert chosen == [4, 5, 6, 7]


def test_should_choose_other_blocks():
    # our server "A" overlaps a crowded region while [4,8) is empty
    spans = {"A": (0, 4, 1.0), "B": (0, 4, 1.0), "C": (0, 8, 1.0)}
    infos = make_infos(8, spans)
    assert block_selection.should_choose_other_blocks("A", infos, balance_quality=0.75)
    # but a balanced swarm stays put
    spans = {"A": (0, 4, 1.0), "B": (4, 8, 1.0)}
    infos = make_infos(8, spans)
    assert not block_selection.should_choose_other_blocks("A", infos, balance_quality=0.75)


def test_allowed_and_blocked_servers():
    """ClientConfig allow/deny lists filter the routing table."""
    spans = {"A": (0, 8, 1.0), "B": (0, 8, 1.0)}
    mgr = _FakeManager(8, spans, allowed_servers=["A"])
    mgr.update()
    for _ in range(4):
        assert [s.peer_id for s in mgr.make_sequence(mode="max_throughput")] == ["A"]
    mgr = _FakeManager(8, spans, blocked_servers=["A"])
    mgr.update()
    for _ in range(4):
        assert [s.peer_id for s in mgr.make_sequence(mode="max_throughput")] == ["B"]
