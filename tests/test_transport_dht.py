import asyncio
import time

import pytest
import torch

from petals_amd.dht.node import DHT, DHTNode
from petals_amd.p2p.transport import P2PNode, RpcError, RpcMessage


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def test_unary_rpc_roundtrip():
    async def main():
        server = P2PNode()

        async def echo(request, stream):
            await stream.close(
                RpcMessage(meta={"echo": request.meta["x"]}, tensors=[t * 2 for t in request.tensors])
            )

        server.add_handler("echo", echo)
        addr = await server.listen()

        client = P2PNode()
        t = torch.randn(3, 4)
        resp = await client.call_unary(addr, "echo", RpcMessage(meta={"x": 42}, tensors=[t]))
        assert resp.meta["echo"] == 42
        assert torch.allclose(resp.tensors[0], t * 2)
        await client.shutdown()
        await server.shutdown()

    run(main())


def test_bidi_stream_rpc():
    async def main():
        server = P2PNode()

        async def accumulate(request, stream):
            total = 0.0
            async for msg in stream:
                if msg.tensors:
                    total += msg.tensors[0].sum().item()
                    await stream.send(RpcMessage(meta={"running": total}))
            await stream.close(RpcMessage(meta={"total": total}))

        server.add_handler("acc", accumulate)
        addr = await server.listen()

        client = P2PNode()
        stream = await client.open_stream(addr, "acc", RpcMessage(meta={}))
        expected = 0.0
        for i in range(3):
            t = torch.ones(2) * i
            expected += t.sum().item()
            await stream.send(RpcMessage(tensors=[t]))
            running = await stream.receive(timeout=5)
            assert abs(running.meta["running"] - expected) < 1e-5
        await stream.close()
        final = await stream.receive(timeout=5)
        assert abs(final.meta["total"] - expected) < 1e-5
        await client.shutdown()
        await server.shutdown()

    run(main())


def test_rpc_error_propagates():
    async def main():
        server = P2PNode()

        async def boom(request, stream):
            raise ValueError("kaboom")

        server.add_handler("boom", boom)
        addr = await server.listen()
        client = P2PNode()
        with pytest.raises(RpcError, match="kaboom"):
            await client.call_unary(addr, "boom", RpcMessage())
        await client.shutdown()
        await server.shutdown()

    run(main())


def test_dht_store_get_across_nodes():
    async def main():
        boot = await DHTNode.create()
        n1 = await DHTNode.create(initial_peers=[boot.listen_addr])
        n2 = await DHTNode.create(initial_peers=[boot.listen_addr])

        exp = time.time() + 60
        await n1.store_many([("model.0", "peerA", {"t": 1.0}, exp), ("model.1", "peerA", {"t": 1.0}, exp)])
        await n2.store_many([("model.1", "peerB", {"t": 2.0}, exp)])

        got = await n2.get_many(["model.0", "model.1", "model.9"])
        assert set(got["model.0"].keys()) == {"peerA"}
        assert set(got["model.1"].keys()) == {"peerA", "peerB"}
        assert got["model.9"] == {}
        assert got["model.1"]["peerB"][0] == {"t": 2.0}

        for n in (n1, n2, boot):
            await n.shutdown()

    run(main())


def test_dht_expiration():
    async def main():
        boot = await DHTNode.create()
        n1 = await DHTNode.create(initial_peers=[boot.listen_addr])
        await n1.store_many([("k", "s", 1, time.time() + 0.2)])
        got = await n1.get_many(["k"])
        assert "s" in got["k"]
        await asyncio.sleep(0.3)
        got = await n1.get_many(["k"])
        assert got["k"] == {}
        await n1.shutdown()
        await boot.shutdown()

    run(main())


def test_threaded_dht_wrapper():
    boot = DHT()
    client = DHT(initial_peers=[boot.listen_addr], client_mode=True)
    exp = time.time() + 30
    boot_peers = boot.store_many([("key", "sub", {"v": 3}, exp)])
    assert boot_peers >= 0
    got = client.get_many(["key"])
    assert got["key"]["sub"][0] == {"v": 3}
    client.shutdown()
    boot.shutdown()
