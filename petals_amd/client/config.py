"""Client-side knobs (parity: reference client/config.py:13-35)."""

from __future__ import annotations

import dataclasses
import os
from typing import Optional, Sequence, Tuple

MAX_RETRIES = int(os.environ.get("PETALS_AMD_MAX_RETRIES", "10"))


@dataclasses.dataclass
class ClientConfig:
    initial_peers: Sequence[Tuple[str, int]] = ()
    dht_prefix: Optional[str] = None

    show_route: str | bool = "inference"  # False / "inference" / True
    allowed_servers: Optional[Sequence[str]] = None
    blocked_servers: Optional[Sequence[str]] = None
    use_server_to_server: bool = True  # rpc_push activation hand-off

    connect_timeout: float = 5.0
    request_timeout: float = 3 * 60.0
    update_period: float = 60.0

    max_retries: Optional[int] = MAX_RETRIES
    min_backoff: float = 1.0
    max_backoff: float = 60.0
    ban_timeout: float = 15.0
    max_pinged: int = 3

    active_adapter: Optional[str] = None

    # transport security (p2p/transport.py STARTTLS): None = inherit the
    # process default / PETALS_AMD_SECURE env; a swarm runs all-secure or
    # all-plain
    secure: Optional[bool] = None

    # wire compression (utils/serialization): "none" | "float16" | "bfloat16" |
    # "blockwise_8bit" — applied to activations we SEND; output_compression is
    # requested from servers for what they send back
    wire_compression: str = "none"
    output_compression: str = "none"
