"""In-process transport shortcut (co-located client+server skips sockets and
serialization) and the LocalMesh ticketed p2p tier (gloo world 2 on CPU; the
same code paths carry RCCL/xGMI traffic on an MI355X node)."""

import asyncio
import os
import socket

import pytest
import torch
import torch.multiprocessing as mp


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


# --------------------------------------------------------------- in-process


def test_inproc_stream_lossless_and_by_reference():
    from petals_amd.p2p.transport import InProcStream, P2PNode, RpcMessage

    async def run():
        node = P2PNode()
        await node.listen(host="127.0.0.1", port=0)
        seen = {}

        async def echo(request, stream):
            seen["tensor"] = request.tensors[0]
            await stream.send(RpcMessage(meta={"ok": True}, tensors=[request.tensors[0] * 2]), kind="end")

        node.add_handler("test.echo", echo)
        t = torch.randn(3, 5, dtype=torch.bfloat16)
        stream = await node.open_stream(node.listen_addr, "test.echo", RpcMessage(tensors=[t]), end=True)
        assert isinstance(stream, InProcStream), "co-located dial must take the in-proc path"
        reply = await stream.receive(timeout=5)
        assert reply.meta["ok"]
        assert reply.tensors[0].dtype == torch.bfloat16
        assert torch.equal(reply.tensors[0], t * 2)
        # lossless + zero-copy: the handler saw the client's tensor object
        assert seen["tensor"] is t
        await node.shutdown()

    asyncio.new_event_loop().run_until_complete(run())


def test_inproc_disabled_falls_back_to_sockets(monkeypatch):
    monkeypatch.setenv("PETALS_AMD_NO_INPROC", "1")
    from petals_amd.p2p.transport import InProcStream, P2PNode, RpcMessage

    async def run():
        node = P2PNode()
        await node.listen(host="127.0.0.1", port=0)

        async def echo(request, stream):
            await stream.send(RpcMessage(meta={"ok": True}, tensors=list(request.tensors)), kind="end")

        node.add_handler("test.echo", echo)
        stream = await node.open_stream(
            node.listen_addr, "test.echo", RpcMessage(tensors=[torch.ones(2)]), end=True
        )
        assert not isinstance(stream, InProcStream)
        reply = await stream.receive(timeout=5)
        assert torch.equal(reply.tensors[0], torch.ones(2))
        await node.shutdown()

    asyncio.new_event_loop().run_until_complete(run())


def test_inproc_error_propagates():
    from petals_amd.p2p.transport import P2PNode, RpcError, RpcMessage

    async def run():
        node = P2PNode()
        await node.listen(host="127.0.0.1", port=0)

        async def boom(request, stream):
            raise ValueError("kaput")

        node.add_handler("test.boom", boom)
        stream = await node.open_stream(node.listen_addr, "test.boom", RpcMessage(), end=True)
        with pytest.raises(RpcError, match="kaput"):
            await stream.receive(timeout=5)
        await node.shutdown()

    asyncio.new_event_loop().run_until_complete(run())


# -------------------------------------------------------------------- mesh


def _mesh_worker(rank, world, port, fail_q):
    try:
        import torch.distributed as dist

        dist.init_process_group(
            backend="gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world
        )
        from petals_amd.parallel.mesh import LocalMesh

        mesh = LocalMesh("test-mesh", rank, world, device=torch.device("cpu"))
        n = 6
        if rank == 0:
            # send n tensors to rank 1; tickets are auto-assigned in order
            futs = []
            for i in range(n):
                fut, ticket = mesh.send(torch.full((2, 3), float(i)), dst=1)
                assert ticket == i
                futs.append(fut)
            for f in futs:
                f.result(timeout=30)
            # bidirectional: also receive one from rank 1 (no deadlock)
            h = mesh.post_recv(src=1, ticket=0, shape=(4,), dtype_str="f32")
            got = h.result(timeout=30)
            assert torch.equal(got, torch.arange(4.0))
        else:
            # post recvs in REVERSE ticket order: the mesh must hold them back
            # and still deliver each ticket's own payload
            handles = {}
            for i in reversed(range(n)):
                handles[i] = mesh.post_recv(src=0, ticket=i, shape=(2, 3), dtype_str="f32")
            fut, _t = mesh.send(torch.arange(4.0), dst=0)
            fut.result(timeout=30)
            for i in range(n):
                got = handles[i].result(timeout=30)
                assert torch.equal(got, torch.full((2, 3), float(i))), (i, got)
        mesh.shutdown()
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def test_mesh_ticket_ordering_world2():
    port = _free_port()
    ctx = mp.get_context("spawn")
    fail_q = ctx.Queue()
    procs = [ctx.Process(target=_mesh_worker, args=(r, 2, port, fail_q)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    errors = []
    while not fail_q.empty():
        errors.append(fail_q.get())
    assert not errors, errors
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
