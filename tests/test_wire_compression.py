"""Wire compression hooks (reference tests/test_remote_sequential.py:65-86):
fp16 / blockwise-8bit compressed activations still produce close outputs."""

import pytest
import torch

from petals_amd.models.config_base import load_model_config


@pytest.fixture(scope="module")
def swarm():
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server

    boot = DHT(host="127.0.0.1")
    server = Server(
        "test-llama", initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
        torch_dtype="float32", block_indices="0:4", dht_prefix="wc-llama", throughput=1.0,
    ).start()
    yield boot
    server.shutdown()
    boot.shutdown()


@pytest.mark.parametrize("compression", ["float16", "blockwise_8bit"])
def test_compressed_forward_close(swarm, compression, monkeypatch):
    # this test exercises the WIRE (lossy compression); the in-process
    # transport shortcut is lossless by design, so force sockets here
    monkeypatch.setenv("PETALS_AMD_NO_INPROC", "1")
    from petals_amd.utils.auto_config import AutoDistributedModel

    boot = swarm
    base = AutoDistributedModel.from_pretrained(
        "test-llama", initial_peers=[boot.listen_addr], dht_prefix="wc-llama",
        show_route=False, max_retries=1,
    )
    torch.manual_seed(0)
    h = torch.randn(1, 6, base.config.hidden_size) * 0.3
    with torch.no_grad():
        ref = base.h(h)
        base.h.sequence_manager.config.wire_compression = compression
        base.h.sequence_manager.config.output_compression = compression
        out = base.h(h)
    atol = 0.05 if compression == "float16" else 0.15
    assert not torch.equal(out, ref) or compression == "float16"
    assert torch.allclose(out, ref, atol=atol, rtol=0.1), (out - ref).abs().max()
    base.h.sequence_manager.shutdown()
