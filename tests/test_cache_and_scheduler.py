"""MemoryCache alloc-timeout/queueing semantics (reference tests/test_cache.py)
and PriorityRuntime ordering (reference tests/test_priority_pool.py)."""

import asyncio
import threading
import time

import pytest
import torch

from petals_amd.server.memory_cache import AllocationFailed, MemoryCache, TensorDescriptor
from petals_amd.server.scheduler import PriorityRuntime


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def desc(n_floats):
    return TensorDescriptor((n_floats,), torch.float32)


def test_alloc_and_free():
    async def main():
        cache = MemoryCache(1024, torch.device("cpu"))
        async with cache.allocate_cache(desc(128), desc(64)) as handles:
            assert len(handles) == 2
            assert cache.current_size_bytes == (128 + 64) * 4
            with cache.use_cache(*handles) as (a, b):
                assert a.shape == (128,) and b.shape == (64,)
                a[:] = 7
            with cache.use_cache(handles[0]) as (a2,):
                assert (a2 == 7).all()
        assert cache.current_size_bytes == 0

    run(main())


def test_alloc_too_big_fails_fast():
    async def main():
        cache = MemoryCache(256, torch.device("cpu"))
        with pytest.raises(AllocationFailed, match="exceeds total cache size"):
            async with cache.allocate_cache(desc(1024)):
                pass

    run(main())


def test_alloc_timeout_when_full():
    async def main():
        cache = MemoryCache(1024, torch.device("cpu"))
        async with cache.allocate_cache(desc(256)):  # fills the cache
            t0 = time.monotonic()
            with pytest.raises(AllocationFailed, match="could not allocate"):
                async with cache.allocate_cache(desc(256), timeout=0.4):
                    pass
            assert 0.3 < time.monotonic() - t0 < 3.0

    run(main())


def test_alloc_waits_for_free():
    async def main():
        cache = MemoryCache(1024, torch.device("cpu"))

        async def holder(release_after):
            async with cache.allocate_cache(desc(256)):
                await asyncio.sleep(release_after)

        hold = asyncio.ensure_future(holder(0.3))
        await asyncio.sleep(0.05)
        t0 = time.monotonic()
        async with cache.allocate_cache(desc(256), timeout=5.0) as handles:
            assert handles
            assert time.monotonic() - t0 > 0.15  # actually waited
        await hold

    run(main())


def test_alloc_fifo_order():
    async def main():
        cache = MemoryCache(1024, torch.device("cpu"))
        order = []

        async def client(name, delay):
            await asyncio.sleep(delay)
            async with cache.allocate_cache(desc(200), timeout=10.0):
                order.append(name)
                await asyncio.sleep(0.1)

        await asyncio.gather(client("a", 0.0), client("b", 0.02), client("c", 0.04))
        assert order == ["a", "b", "c"]

    run(main())


def test_priority_runtime_ordering():
    runtime = PriorityRuntime(torch.device("cpu")).start()
    gate = threading.Event()
    order = []

    async def main():
        def blocker():
            gate.wait(5)
            order.append("blocker")

        def training():
            order.append("training")

        def inference():
            order.append("inference")

        t_block = asyncio.ensure_future(runtime.submit(0.5, blocker))
        await asyncio.sleep(0.1)  # blocker is running; queue the rest
        t_train = asyncio.ensure_future(runtime.submit(2.0, training))
        t_inf = asyncio.ensure_future(runtime.submit(1.0, inference))
        await asyncio.sleep(0.05)
        gate.set()
        await asyncio.gather(t_block, t_train, t_inf)

    run(main())
    runtime.shutdown()
    assert order == ["blocker", "inference", "training"]  # priority 1.0 beats 2.0


def test_priority_runtime_exception_propagates():
    runtime = PriorityRuntime(torch.device("cpu")).start()

    async def main():
        def boom():
            raise ValueError("pow")

        with pytest.raises(ValueError, match="pow"):
            await runtime.submit(1.0, boom)

    run(main())
    runtime.shutdown()
