"""Priority runtime: a single thread that owns the GPU and executes compute
tasks in priority order (inference beats training).

Replaces the reference's hivemind Runtime + PrioritizedTaskPool machinery
(`server/task_pool.py`, `server/server.py:770-775`): there, N handler
*processes* feed a runtime *process* through shared-memory futures; here,
asyncio handlers and the runtime thread share one process, so a task is just
a closure and an asyncio future. Priority and task-size accounting semantics
are preserved (`server/task_pool.py:29-177`).

On GPU, inference tasks run on a dedicated high-priority HIP stream and
training tasks on a second stream, so a long training forward does not add
latency to token decoding beyond the kernel in flight.
"""

from __future__ import annotations

import asyncio
import heapq
import itertools
import logging
import threading
import time
from typing import Any, Callable, Optional

import torch

logger = logging.getLogger(__name__)

import os as _os

_TRACE = bool(_os.environ.get("PETALS_AMD_STEP_TRACE"))


class TaskPrioritizer:
    """Policy hook (parity: server/task_prioritizer.py:15). Lower = sooner."""

    def prioritize(self, *tensors, points: float = 0.0, type: str = "inference", **kwargs) -> float:
        if type == "inference":
            return 1.0
        return 2.0


class PriorityRuntime:
    def __init__(self, device: Optional[torch.device] = None, sync_every: bool = True):
        if device is not None:
            device = torch.device(device)
            if device.type == "cuda" and device.index is None:
                device = torch.device("cuda", torch.cuda.current_device() if torch.cuda.is_available() else 0)
        self.device = device
        self._queue: list = []
        self._cv = threading.Condition()
        self._counter = itertools.count()
        self._shutdown = False
        self._thread: Optional[threading.Thread] = None
        self._streams = {}
        self.stats = {"tasks": 0, "busy_time": 0.0}
        # keep-warm (opt-in, PETALS_AMD_KEEP_WARM=<seconds>): for a short
        # window after each task, keep the device queue non-empty while idle.
        # Per-token serving leaves ~10 ms host gaps in which this pool's host
        # power management parks sclk, making ~every 3rd step run ~5x slower
        # (PARITY.md "serving cadence"). 1-element ticks every 1 ms were NOT
        # enough to hold clocks on the SR-IOV pool host; the knob stays for
        # bare-metal deployments where guest clocks are controllable.
        self._keep_warm_s = float(_os.environ.get("PETALS_AMD_KEEP_WARM", "0"))
        self._warm_until = 0.0
        self._warm_buf: Optional[torch.Tensor] = None

    def start(self):
        self._thread = threading.Thread(target=self._run, name="PriorityRuntime", daemon=True)
        self._thread.start()
        return self

    def _get_stream(self, priority: float):
        if self.device is None or self.device.type != "cuda":
            return None
        key = "inference" if priority <= 1.0 else "training"
        if key not in self._streams:
            self._streams[key] = torch.cuda.Stream(
                device=self.device, priority=-1 if key == "inference" else 0
            )
        return self._streams[key]

    def _run(self):
        if self.device is not None and self.device.type == "cuda":
            torch.cuda.set_device(self.device)
        while True:
            with self._cv:
                while not self._queue and not self._shutdown:
                    if (
                        self.device is not None
                        and self.device.type == "cuda"
                        and self._keep_warm_s > 0
                        and time.monotonic() < self._warm_until
                    ):
                        self._warm_tick()
                        self._cv.wait(timeout=0.001)
                    else:
                        self._cv.wait(timeout=1.0)
                if self._shutdown and not self._queue:
                    return
                if not self._queue:
                    continue
                priority, _seq, fn, args, kwargs, loop, future = heapq.heappop(self._queue)
            t0 = time.perf_counter()
            if _TRACE and isinstance(kwargs.get("_enq"), float):
                print(f"[rt] queue-wait {(t0 - kwargs.pop('_enq'))*1e3:.2f} ms", flush=True)
            else:
                kwargs.pop("_enq", None)
            try:
                stream = self._get_stream(priority)
                if stream is not None:
                    with torch.cuda.stream(stream):
                        result = fn(*args, **kwargs)
                    stream.synchronize()
                else:
                    result = fn(*args, **kwargs)
                loop.call_soon_threadsafe(_set_result_safe, future, result)
            except BaseException as e:  # noqa: BLE001
                loop.call_soon_threadsafe(_set_exception_safe, future, e)
            finally:
                self._warm_until = time.monotonic() + self._keep_warm_s
                self.stats["tasks"] += 1
                if _TRACE:
                    print(f"[rt] task {(time.perf_counter()-t0)*1e3:.2f} ms", flush=True)
                self.stats["busy_time"] += time.perf_counter() - t0

    def _warm_tick(self):
        """Keep the device queue non-empty during the keep-warm window: a
        ~1 ms bounded spin kernel (ops warm_spin, 1 workgroup, s_sleep loop)
        if the HIP extension is loaded, else a trivial add."""
        if self._warm_buf is None:
            self._warm_buf = torch.zeros(1, device=self.device)
            from petals_amd import ops as _ops

            self._warm_spin = getattr(_ops._load_hip_ops(), "warm_spin", None)
        stream = self._get_stream(1.0)
        with torch.cuda.stream(stream):
            if self._warm_spin is not None:
                self._warm_spin(1000)
            else:
                self._warm_buf.add_(0.0)

    async def submit(self, priority: float, fn: Callable, *args, **kwargs) -> Any:
        """Schedule fn on the runtime thread; await its result."""
        if _TRACE:
            kwargs["_enq"] = time.perf_counter()
        loop = asyncio.get_event_loop()
        future: asyncio.Future = loop.create_future()
        with self._cv:
            if self._shutdown:
                raise RuntimeError("runtime is shut down")
            heapq.heappush(self._queue, (priority, next(self._counter), fn, args, kwargs, loop, future))
            self._cv.notify()
        return await future

    def submit_sync(self, priority: float, fn: Callable, *args, timeout: float = 600.0, **kwargs) -> Any:
        """Blocking submit from a non-asyncio thread."""
        done = threading.Event()
        box: dict = {}

        def wrapper():
            try:
                box["result"] = fn(*args, **kwargs)
            except BaseException as e:  # noqa: BLE001
                box["error"] = e
            finally:
                done.set()

        with self._cv:
            if self._shutdown:
                raise RuntimeError("runtime is shut down")
            heapq.heappush(
                self._queue,
                (priority, next(self._counter), wrapper, (), {}, _DummyLoop(), _DummyFuture()),
            )
            self._cv.notify()
        if not done.wait(timeout):
            raise TimeoutError("runtime task timed out")
        if "error" in box:
            raise box["error"]
        return box["result"]

    @property
    def alive(self) -> bool:
        return self._thread is not None and self._thread.is_alive()

    def shutdown(self):
        with self._cv:
            self._shutdown = True
            self._cv.notify_all()
        if self._thread is not None:
            self._thread.join(timeout=10)


def _set_result_safe(future: asyncio.Future, result):
    if not future.done():
        future.set_result(result)


def _set_exception_safe(future: asyncio.Future, exc):
    if not future.done():
        future.set_exception(exc)


class _DummyLoop:
    def call_soon_threadsafe(self, fn, *args):
        fn(*args)


class _DummyFuture:
    def done(self):
        return True
