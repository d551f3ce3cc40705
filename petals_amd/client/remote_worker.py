"""A dedicated asyncio worker thread for client-side RPC coroutines.

Parity with hivemind's RemoteExpertWorker (used at reference
client/sequential_autograd.py:239): RPC work must NOT run on the DHT's event
loop — routing code called from inside an RPC coroutine blocks on DHT lookups,
which would deadlock a shared loop. The worker owns its own P2PNode (outbound
connections only).
"""

from __future__ import annotations

import asyncio
import threading
from typing import Optional

from petals_amd.p2p.transport import P2PNode


class RemoteWorker:
    def __init__(self):
        self.loop: Optional[asyncio.AbstractEventLoop] = None
        self.p2p: Optional[P2PNode] = None
        self._thread: Optional[threading.Thread] = None
        self._ready = threading.Event()
        self._lock = threading.Lock()

    def _ensure_started(self):
        with self._lock:
            if self._thread is not None and self._thread.is_alive():
                return
            self._ready.clear()
            self._thread = threading.Thread(target=self._run, name="RemoteWorker", daemon=True)
            self._thread.start()
        self._ready.wait(timeout=10)

    def _run(self):
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
        self.loop = loop
        self.p2p = P2PNode()
        self._ready.set()
        loop.run_forever()

    def run_coroutine(self, coro, timeout: Optional[float] = None):
        self._ensure_started()
        future = asyncio.run_coroutine_threadsafe(coro, self.loop)
        try:
            return future.result(timeout)
        except BaseException:
            future.cancel()
            raise

    def shutdown(self):
        if self.loop is not None:
            self.loop.call_soon_threadsafe(self.loop.stop)
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None


_singleton: Optional[RemoteWorker] = None
_singleton_lock = threading.Lock()


def get_worker() -> RemoteWorker:
    global _singleton
    with _singleton_lock:
        if _singleton is None:
            _singleton = RemoteWorker()
        return _singleton


def reset_worker() -> None:
    """Tear down the shared worker (its P2PNode captured the process's
    security mode at first use — call this when flipping modes, e.g. tests)."""
    global _singleton
    with _singleton_lock:
        if _singleton is not None:
            _singleton.shutdown()
            _singleton = None
