"""Parallel RTT pings with EMA smoothing (parity: reference utils/ping.py:40)."""

from __future__ import annotations

import asyncio
import math
import time
from typing import Dict, Tuple

from petals_amd.p2p.transport import P2PNode, RpcMessage


class PingAggregator:
    def __init__(self, p2p: P2PNode, *, ema_alpha: float = 0.2, expiration: float = 300.0):
        self.p2p = p2p
        self.ema_alpha = ema_alpha
        self.expiration = expiration
        self.ping_emas: Dict[str, Tuple[float, float]] = {}  # peer -> (rtt_ema, ts)

    async def ping(self, peers_with_addrs: Dict[str, Tuple[str, int]], wait_timeout: float = 5.0) -> None:
        async def one(peer_id: str, addr):
            t0 = time.perf_counter()
            try:
                await asyncio.wait_for(
                    self.p2p.call_unary(tuple(addr), "dht.ping", RpcMessage(meta={})), wait_timeout
                )
                rtt = time.perf_counter() - t0
            except Exception:  # noqa: BLE001
                rtt = math.inf
            old = self.ping_emas.get(peer_id)
            if old is not None and math.isfinite(old[0]) and math.isfinite(rtt):
                rtt = self.ema_alpha * rtt + (1 - self.ema_alpha) * old[0]
            self.ping_emas[peer_id] = (rtt, time.monotonic())

        await asyncio.gather(*(one(p, a) for p, a in peers_with_addrs.items()))

    def to_dict(self) -> Dict[str, float]:
        now = time.monotonic()
        return {p: rtt for p, (rtt, ts) in self.ping_emas.items() if now - ts < self.expiration}
