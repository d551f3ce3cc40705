"""Transport hardening: STARTTLS with keypair-bound peer ids, impersonation
rejection, and the circuit-relay fallback (rendezvous + byte splice)."""

import asyncio

import pytest
import torch

from petals_amd.p2p.identity import NodeIdentity, cert_fingerprint
from petals_amd.p2p.relay import RelayClient, RelayHub
from petals_amd.p2p.transport import InProcStream, P2PNode, RpcError, RpcMessage


def _run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def test_identity_peer_id_is_cert_fingerprint(tmp_path):
    ident = NodeIdentity(str(tmp_path / "id"))
    assert ident.peer_id == cert_fingerprint(ident.cert_der)
    # reloading the same dir keeps the identity stable
    ident2 = NodeIdentity(str(tmp_path / "id"))
    assert ident2.peer_id == ident.peer_id


def test_tls_roundtrip_and_encryption(monkeypatch, tmp_path):
    monkeypatch.setenv("PETALS_AMD_NO_INPROC", "1")  # exercise real sockets

    async def run():
        server = P2PNode(secure=True, identity=NodeIdentity(str(tmp_path / "srv")))
        await server.listen(host="127.0.0.1", port=0)

        async def echo(request, stream):
            await stream.send(RpcMessage(meta={"ok": True}, tensors=list(request.tensors)), kind="end")

        server.add_handler("t.echo", echo)

        client = P2PNode(secure=True)
        client.expect_peer(server.listen_addr, server.peer_id)
        t = torch.randn(3, 4)
        reply = await client.call_unary(server.listen_addr, "t.echo", RpcMessage(tensors=[t]))
        assert reply.meta["ok"] and torch.allclose(reply.tensors[0], t)
        # the connection really is TLS
        conn = next(iter(client._conns.values()))
        assert conn.writer.transport.get_extra_info("ssl_object") is not None
        await client.shutdown()
        await server.shutdown()

    _run(run())


def test_tls_rejects_impersonation(monkeypatch, tmp_path):
    monkeypatch.setenv("PETALS_AMD_NO_INPROC", "1")

    async def run():
        server = P2PNode(secure=True, identity=NodeIdentity(str(tmp_path / "srv")))
        await server.listen(host="127.0.0.1", port=0)
        client = P2PNode(secure=True)
        # the DHT claims a different peer id for this address
        client.expect_peer(server.listen_addr, "f" * 32)
        with pytest.raises(RpcError, match="identity mismatch"):
            await client.call_unary(server.listen_addr, "t.none", RpcMessage())
        await client.shutdown()
        await server.shutdown()

    _run(run())


def test_secure_server_refuses_plaintext(monkeypatch, tmp_path):
    monkeypatch.setenv("PETALS_AMD_NO_INPROC", "1")

    async def run():
        server = P2PNode(secure=True, identity=NodeIdentity(str(tmp_path / "srv")))
        await server.listen(host="127.0.0.1", port=0)
        client = P2PNode(secure=False)
        with pytest.raises((RpcError, asyncio.TimeoutError, OSError)):
            await asyncio.wait_for(
                client.call_unary(server.listen_addr, "t.none", RpcMessage(), timeout=3), 5
            )
        await client.shutdown()
        await server.shutdown()

    _run(run())


@pytest.fixture(scope="module")
def tiny_ckpt(tmp_path_factory):
    transformers = pytest.importorskip("transformers")
    torch.manual_seed(0)
    cfg = transformers.LlamaConfig(
        hidden_size=64, num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        intermediate_size=128, vocab_size=128, max_position_embeddings=256,
        tie_word_embeddings=False,
    )
    model = transformers.LlamaForCausalLM(cfg).eval()
    path = tmp_path_factory.mktemp("sec_ckpt")
    model.save_pretrained(path, safe_serialization=True)
    return str(path), model


@pytest.fixture()
def fresh_worker():
    from petals_amd.client.remote_worker import reset_worker

    reset_worker()
    yield
    reset_worker()


def test_secure_swarm_generate_exact_match(monkeypatch, tiny_ckpt, fresh_worker):
    """A fully TLS swarm (boot DHT + servers + client) serves generate() with
    outputs identical to local HF."""
    monkeypatch.setenv("PETALS_AMD_SECURE", "1")
    monkeypatch.setenv("PETALS_AMD_NO_INPROC", "1")  # force the TLS sockets
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    path, hf_model = tiny_ckpt
    boot = DHT(host="127.0.0.1")
    servers = [
        Server(path, initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
               torch_dtype="float32", block_indices=spec, dht_prefix="sec-swarm",
               throughput=1.0).start()
        for spec in ("0:2", "2:4")
    ]
    try:
        model = AutoDistributedModelForCausalLM.from_pretrained(
            path, initial_peers=[boot.listen_addr], dht_prefix="sec-swarm",
            show_route=False, max_retries=1, min_backoff=0.2,
        )
        torch.manual_seed(21)
        ids = torch.randint(0, 128, (1, 5))
        ref = hf_model.generate(ids, max_new_tokens=6, do_sample=False)
        out = model.generate(ids, max_new_tokens=6, do_sample=False)
        assert torch.equal(out, ref), (out, ref)
        # the servers' peer ids really are certificate fingerprints
        assert all(len(s.peer_id) == 32 for s in servers)
        model.transformer.h.sequence_manager.shutdown()
    finally:
        for s in servers:
            s.shutdown()
        boot.shutdown()


def test_relayed_server_serves_blocks(monkeypatch, tiny_ckpt, fresh_worker):
    """A server with force_relay=True announces a relayed address, serves
    through the bootstrap node's splice, and gets the x0.2 throughput penalty."""
    monkeypatch.setenv("PETALS_AMD_NO_INPROC", "1")
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    path, hf_model = tiny_ckpt
    boot = DHT(host="127.0.0.1")
    s1 = Server(path, initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
                torch_dtype="float32", block_indices="0:2", dht_prefix="relay-swarm",
                throughput=10.0).start()
    s2 = Server(path, initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
                torch_dtype="float32", block_indices="2:4", dht_prefix="relay-swarm",
                throughput=10.0, force_relay=True).start()
    try:
        assert s2._relayed and s2.server_info.using_relay
        assert s2.server_info.throughput == pytest.approx(2.0)  # 10 x 0.2
        model = AutoDistributedModelForCausalLM.from_pretrained(
            path, initial_peers=[boot.listen_addr], dht_prefix="relay-swarm",
            show_route=False, max_retries=1, min_backoff=0.2,
        )
        torch.manual_seed(22)
        ids = torch.randint(0, 128, (1, 5))
        ref = hf_model.generate(ids, max_new_tokens=6, do_sample=False)
        out = model.generate(ids, max_new_tokens=6, do_sample=False)
        assert torch.equal(out, ref), (out, ref)
        model.transformer.h.sequence_manager.shutdown()
    finally:
        s1.shutdown()
        s2.shutdown()
        boot.shutdown()


@pytest.mark.parametrize("secure", [False, True])
def test_relay_tunnel_end_to_end(monkeypatch, tmp_path, secure):
    """An 'unreachable' node serves through a relay; with secure=True the TLS
    session terminates at the target (the relay only splices bytes)."""
    monkeypatch.setenv("PETALS_AMD_NO_INPROC", "1")

    async def run():
        relay = P2PNode(secure=secure, identity=NodeIdentity(str(tmp_path / "relay")) if secure else None)
        await relay.listen(host="127.0.0.1", port=0)
        RelayHub(relay)

        target = P2PNode(secure=secure, identity=NodeIdentity(str(tmp_path / "target")) if secure else None)
        # NB: target NEVER listens — it is "unreachable"

        async def whoami(request, stream):
            await stream.send(RpcMessage(meta={"peer": target.peer_id}), kind="end")

        target.add_handler("t.whoami", whoami)
        rc = RelayClient(target, relay.listen_addr)
        await rc.start()
        await asyncio.sleep(0.1)

        client = P2PNode(secure=secure)
        relayed_addr = (relay.listen_addr[0], relay.listen_addr[1], "relay", target.peer_id)
        reply = await client.call_unary(relayed_addr, "t.whoami", RpcMessage(), timeout=15)
        assert reply.meta["peer"] == target.peer_id
        if secure:
            conn = next(iter(client._conns.values()))
            ssl_obj = conn.writer.transport.get_extra_info("ssl_object")
            assert ssl_obj is not None
            # end-to-end: the certificate is the TARGET's, not the relay's
            assert cert_fingerprint(ssl_obj.getpeercert(binary_form=True)) == target.peer_id
        await client.shutdown()
        await target.shutdown()
        await relay.shutdown()

    _run(run())
