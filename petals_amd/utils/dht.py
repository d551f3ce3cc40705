"""Block-announcement helpers over the DHT.

Parity with reference ``utils/dht.py``: declare_active_modules (:28),
get_remote_module_infos (:74), compute_spans (:134). Key = module uid,
subkey = peer_id hex, value = {"info": ServerInfo dict, "addr": [host, port]}.
Unlike the reference (where the DHT resolves peer_id -> multiaddr via libp2p),
we gossip the server's TCP address together with its info.
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional, Sequence, Tuple

from petals_amd.data_structures import (
    ModuleUID,
    RemoteModuleInfo,
    ServerInfo,
    compute_spans,  # re-export for callers
)
from petals_amd.dht.node import DHT

__all__ = ["declare_active_modules", "get_remote_module_infos", "compute_spans", "declare_model", "list_models"]

MODELS_KEY = "_petals_amd.models"


def declare_active_modules(
    dht: DHT,
    uids: Sequence[ModuleUID],
    server_info: ServerInfo,
    server_addr: Tuple[str, int],
    expiration_time: float,
) -> int:
    entries = []
    value = {"info": server_info.to_dict(), "addr": [server_addr[0], server_addr[1]]}
    for uid in uids:
        entries.append((uid, dht.peer_id, value, expiration_time))
    return dht.store_many(entries)


def get_remote_module_infos(
    dht: DHT,
    uids: Sequence[ModuleUID],
    *,
    active_only: bool = True,
) -> Tuple[List[Optional[RemoteModuleInfo]], Dict[str, Tuple[str, int]]]:
    """Returns (module_infos aligned with uids, {peer_id: (host, port)})."""
    found = dht.get_many(list(uids))
    infos: List[Optional[RemoteModuleInfo]] = []
    addrs: Dict[str, Tuple[str, int]] = {}
    for uid in uids:
        servers: Dict[str, ServerInfo] = {}
        for peer_id, (value, _expiration) in (found.get(uid) or {}).items():
            try:
                info = ServerInfo.from_dict(value["info"])
                # addr may be [host, port] or [relay_host, relay_port, "relay",
                # target_peer_id] for NAT'd servers behind a circuit relay
                addr = (value["addr"][0], int(value["addr"][1]), *value["addr"][2:])
            except (KeyError, TypeError, ValueError):
                continue
            if active_only and info.state != 2:  # ServerState.ONLINE
                continue
            servers[peer_id] = info
            addrs[peer_id] = addr
        infos.append(RemoteModuleInfo(uid=uid, servers=servers) if servers else None)
    return infos, addrs


def declare_model(dht: DHT, model_id: str, info: dict, expiration_time: Optional[float] = None) -> int:
    """Announce that this swarm serves `model_id` (parity: server.py:739-744)."""
    if expiration_time is None:
        expiration_time = time.time() + 300
    return dht.store_many([(MODELS_KEY, model_id, info, expiration_time)])


def list_models(dht: DHT) -> Dict[str, dict]:
    found = dht.get_many([MODELS_KEY]).get(MODELS_KEY) or {}
    return {model_id: value for model_id, (value, _exp) in found.items()}
