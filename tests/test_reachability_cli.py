"""CPU tests for the reachability dial-back protocol and the run_server CLI
flag/YAML handling (SURVEY §5.6 / §2.1 reachability rows)."""

import asyncio

import pytest

from petals_amd.p2p.transport import P2PNode
from petals_amd.server.reachability import (ReachabilityProtocol, check_direct_reachability,
                                            validate_reachability)


def test_reachability_dialback():
    async def run():
        peer = P2PNode()
        await peer.listen("127.0.0.1")
        ReachabilityProtocol(peer)
        me = P2PNode()
        await me.listen("127.0.0.1")
        try:
            # reachable: probe peer can dial our listening address back
            ok = await check_direct_reachability(me, me.listen_addr, [peer.listen_addr])
            assert ok is True
            # unreachable: nothing listens on this port
            ok = await check_direct_reachability(me, ("127.0.0.1", 1), [peer.listen_addr])
            assert ok is False
            # no probes at all -> None (lone server continues)
            ok = await check_direct_reachability(me, me.listen_addr, [])
            assert ok is None
            # validate: passes quickly when reachable
            await validate_reachability(me, me.listen_addr, [peer.listen_addr], wait_time=5)
            # validate: raises when provably unreachable
            with pytest.raises(RuntimeError):
                await validate_reachability(
                    me, ("127.0.0.1", 1), [peer.listen_addr], wait_time=0.1, retry_time=0.05
                )
        finally:
            await me.shutdown()
            await peer.shutdown()

    asyncio.run(run())


def test_run_server_yaml_defaults(tmp_path, monkeypatch):
    """--config YAML fills flag defaults without overriding explicit flags."""
    import petals_amd.cli.run_server as rs

    cfg = tmp_path / "config.yml"
    cfg.write_text("quant_type: nf4\nmax_batch_size: 4\nhost: 1.2.3.4\n")

    captured = {}

    class FakeServer:
        def __init__(self, model, **kwargs):
            captured["model"] = model
            captured.update(kwargs)

        listen_addr = ("0.0.0.0", 0)
        peer_id = "fake"

        def start(self):
            return self

        def is_healthy(self):
            return False  # exits the run loop immediately

        def shutdown(self):
            pass

    monkeypatch.setattr("petals_amd.server.server.Server", FakeServer)
    rs.main(["test-llama", "--config", str(cfg), "--host", "9.9.9.9"])
    assert captured["model"] == "test-llama"
    assert captured["quant_type"] == "nf4"  # from YAML
    assert captured["max_batch_size"] == 4  # from YAML
    assert captured["host"] == "9.9.9.9"  # explicit flag wins over YAML
