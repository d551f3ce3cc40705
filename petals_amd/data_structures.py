"""Shared schema for the swarm: module UIDs, server info gossiped via the DHT,
span descriptors used by routing, and per-request inference metadata.

Capability parity with the reference's ``src/petals/data_structures.py``
(ModuleUID/parse_uid :9-17, ServerState :33, ServerInfo :42-74,
RemoteModuleInfo :77, RemoteSpanInfo :85, InferenceMetadata :112) — the wire
format here is our own (msgpack-friendly plain tuples/dicts).
"""

from __future__ import annotations

import dataclasses
import time
from enum import IntEnum
from typing import Any, Dict, List, Optional, Sequence, Tuple

# A module UID is "{dht_prefix}{UID_DELIMITER}{block_index}", e.g. "llama70b.17"
ModuleUID = str
UID_DELIMITER = "."
CHAIN_DELIMITER = " "  # joins multiple module uids in one RPC target


def make_uid(dht_prefix: str, index: int) -> ModuleUID:
    return f"{dht_prefix}{UID_DELIMITER}{index}"


def parse_uid(uid: ModuleUID) -> Tuple[str, int]:
    assert CHAIN_DELIMITER not in uid, "parse_uid() accepts a single uid"
    dht_prefix, index = uid.rsplit(UID_DELIMITER, 1)
    return dht_prefix, int(index)


class ServerState(IntEnum):
    OFFLINE = 0
    JOINING = 1
    ONLINE = 2


RPS = float


@dataclasses.dataclass
class ServerInfo:
    """Everything a server gossips about itself via the DHT."""

    state: ServerState
    throughput: RPS  # min(compute, network) tokens/sec per block — used by routing

    start_block: Optional[int] = None
    end_block: Optional[int] = None

    public_name: Optional[str] = None
    version: Optional[str] = None

    network_rps: Optional[RPS] = None
    forward_rps: Optional[RPS] = None
    inference_rps: Optional[RPS] = None

    adapters: Sequence[str] = ()
    torch_dtype: Optional[str] = None
    quant_type: Optional[str] = None
    using_relay: bool = False
    cache_tokens_left: Optional[int] = None
    next_pings: Optional[Dict[str, float]] = None  # peer_id hex -> rtt seconds
    # co-located RCCL/xGMI tier (parallel/mesh.py): servers sharing a mesh
    # exchange activations over RCCL p2p instead of TCP
    mesh_id: Optional[str] = None
    mesh_rank: Optional[int] = None

    def to_dict(self) -> Dict[str, Any]:
        d = dataclasses.asdict(self)
        d["state"] = int(self.state)
        d["adapters"] = list(self.adapters)
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ServerInfo":
        d = dict(d)
        known = {f.name for f in dataclasses.fields(cls)}
        d = {k: v for k, v in d.items() if k in known}
        d["state"] = ServerState(d.get("state", 0))
        d["adapters"] = tuple(d.get("adapters", ()))
        return cls(**d)


@dataclasses.dataclass
class RemoteModuleInfo:
    """A single block's uid and the servers that host it right now."""

    uid: ModuleUID
    servers: Dict[str, ServerInfo]  # peer_id hex -> info


@dataclasses.dataclass
class RemoteSpanInfo:
    """A contiguous interval of blocks [start, end) hosted by one server."""

    peer_id: str  # hex peer id
    start: int
    end: int
    server_info: ServerInfo

    def contains(self, block_index: int) -> bool:
        return self.start <= block_index < self.end

    @property
    def length(self) -> int:
        return self.end - self.start

    @property
    def state(self) -> ServerState:
        return self.server_info.state

    @property
    def throughput(self) -> float:
        return self.server_info.throughput


def compute_spans(module_infos: List[Optional[RemoteModuleInfo]]) -> Dict[str, RemoteSpanInfo]:
    """Reconstruct per-server contiguous spans from per-block server maps.

    Parity: reference ``utils/dht.py:compute_spans`` (:134).
    """
    spans: Dict[str, RemoteSpanInfo] = {}
    for block_idx, info in enumerate(module_infos):
        if info is None:
            continue
        for peer_id, server_info in info.servers.items():
            if peer_id in spans and spans[peer_id].end == block_idx:
                spans[peer_id].end = block_idx + 1
            elif peer_id not in spans:
                spans[peer_id] = RemoteSpanInfo(
                    peer_id=peer_id, start=block_idx, end=block_idx + 1, server_info=server_info
                )
            else:
                # non-contiguous announcement: keep the longer span
                if spans[peer_id].length < 1:
                    spans[peer_id] = RemoteSpanInfo(
                        peer_id=peer_id, start=block_idx, end=block_idx + 1, server_info=server_info
                    )
    return spans


@dataclasses.dataclass(frozen=True)
class InferenceMetadata:
    """Metadata shipped alongside each inference step inside the server."""

    uid: ModuleUID
    prefix_length: int
    cache_handles: Tuple[int, ...]
    active_adapter: Optional[str] = None


def get_dht_time() -> float:
    return time.time()
