"""KV-cache allocator for one server process.

Parity with reference ``server/memory_cache.py`` (allocate_cache :71,
_wait_for_free_memory :136, use_cache :195, AllocationFailed :224) with one
big simplification enabled by the MI355X-native single-process design: the
reference splits handlers (N subprocesses) from the GPU runtime (1 process)
and moves tensors through shared memory + pipes; we run asyncio handlers and
a single runtime thread in ONE process, so a handle maps directly to a live
tensor. Alloc-timeout + FIFO queueing semantics are preserved because client
routing depends on them (`cache_tokens_left` is a routing edge cost).
"""

from __future__ import annotations

import asyncio
import contextlib
import itertools
import logging
import threading
import time
from typing import Dict, List, Optional, Tuple

import torch

from petals_amd.utils.misc import get_size_in_bytes

logger = logging.getLogger(__name__)

Handle = int


class AllocationFailed(Exception):
    pass


class TensorDescriptor:
    __slots__ = ("shape", "dtype")

    def __init__(self, shape: Tuple[int, ...], dtype: torch.dtype):
        self.shape = tuple(shape)
        self.dtype = dtype

    @property
    def numel(self) -> int:
        n = 1
        for s in self.shape:
            n *= s
        return n

    @property
    def nbytes(self) -> int:
        return self.numel * get_size_in_bytes(self.dtype)


class MemoryCache:
    """Byte-budgeted allocator of per-session KV tensors on one device."""

    def __init__(self, max_size_bytes: Optional[int], device: torch.device, alloc_timeout: float = 600.0):
        self.max_size_bytes = max_size_bytes if max_size_bytes is not None else (1 << 62)
        self.device = torch.device(device)
        self.alloc_timeout = alloc_timeout
        self._lock = threading.Lock()
        self._current_size = 0
        self._handle_counter = itertools.count()
        self._tensors: Dict[Handle, torch.Tensor] = {}
        self._freed_event = asyncio.Event()
        self._alloc_queue: List[int] = []  # FIFO ticket queue for fairness
        self._ticket_counter = itertools.count()

    @property
    def current_size_bytes(self) -> int:
        return self._current_size

    @property
    def bytes_left(self) -> int:
        return max(0, self.max_size_bytes - self._current_size)

    @contextlib.asynccontextmanager
    async def allocate_cache(
        self, *descriptors: TensorDescriptor, timeout: Optional[float] = None
    ):
        """Async context manager: reserves memory (waiting FIFO up to `timeout`
        seconds if the cache is full), yields handles, frees on exit."""
        if timeout is None:
            timeout = self.alloc_timeout
        total_bytes = sum(d.nbytes for d in descriptors)
        if total_bytes > self.max_size_bytes:
            raise AllocationFailed(
                f"request of {total_bytes} bytes exceeds total cache size {self.max_size_bytes}"
            )
        handles: Optional[List[Handle]] = None
        ticket = next(self._ticket_counter)
        self._alloc_queue.append(ticket)
        deadline = time.monotonic() + timeout
        try:
            while True:
                with self._lock:
                    is_turn = self._alloc_queue and self._alloc_queue[0] == ticket
                    if is_turn and self._current_size + total_bytes <= self.max_size_bytes:
                        handles = []
                        for d in descriptors:
                            handle = next(self._handle_counter)
                            self._tensors[handle] = torch.zeros(d.shape, dtype=d.dtype, device=self.device)
                            handles.append(handle)
                        self._current_size += total_bytes
                        self._alloc_queue.remove(ticket)
                        break
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    raise AllocationFailed(
                        f"could not allocate {total_bytes} bytes within {timeout} s "
                        f"(used {self._current_size}/{self.max_size_bytes})"
                    )
                self._freed_event.clear()
                try:
                    await asyncio.wait_for(self._freed_event.wait(), min(remaining, 1.0))
                except asyncio.TimeoutError:
                    pass
            yield tuple(handles)
        finally:
            if ticket in self._alloc_queue:
                self._alloc_queue.remove(ticket)
            if handles is not None:
                with self._lock:
                    for handle in handles:
                        t = self._tensors.pop(handle, None)
                        if t is not None:
                            self._current_size -= t.numel() * get_size_in_bytes(t.dtype)
                self._freed_event.set()

    @contextlib.contextmanager
    def use_cache(self, *handles: Handle):
        """Returns live cache tensors for the runtime thread."""
        tensors = []
        with self._lock:
            for handle in handles:
                if handle not in self._tensors:
                    raise KeyError(f"unknown cache handle {handle}")
                tensors.append(self._tensors[handle])
        yield tuple(tensors)
