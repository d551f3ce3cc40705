"""Tensor-parallel shard equivalence vs a single full block, world_size=2 on
CPU/gloo (parity: reference tests/test_tensor_parallel.py — its CI runs
`--tensor_parallel_devices cpu cpu`)."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _tp_worker(rank, world, port, payload_path, result_path):
    import torch
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from petals_amd.models.config_base import load_model_config
        from petals_amd.parallel.tp import build_tp_block

        payload = torch.load(payload_path, weights_only=False)
        cfg = load_model_config(payload.get("model", "test-llama"))
        block = build_tp_block(cfg, 0, rank=rank, world=world)
        block.load_from_full_state_dict(payload["state_dict"])
        x = payload["x"]

        # full forward
        out = block(x)
        # incremental with per-rank shard caches
        ks, vs = block.kv_cache_shape(x.shape[0], 16)
        k, v = torch.zeros(ks), torch.zeros(vs)
        parts = [block(x[:, :3], kv_cache=(k, v), prefix_length=0)]
        for t in range(3, x.shape[1]):
            parts.append(block(x[:, t : t + 1], kv_cache=(k, v), prefix_length=t))
        inc = torch.cat(parts, 1)

        # training backward: grads wrt the replicated input must equal the
        # full block's (copy_to_tp all-reduces input grads across shards)
        xg = x.detach().clone().requires_grad_(True)
        block(xg).square().mean().backward()

        if rank == 0:
            torch.save({"out": out, "inc": inc, "x_grad": xg.grad}, result_path)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize(
    "model",
    [
        "test-llama",
        "test-falcon",  # new-decoder GQA (single-reduce parallel residual)
        "test-falcon-mqa",  # MQA with a PRIME head count (uneven shards, replicated kv)
        "test-falcon-classic",  # sequential residual + biased linears (rank-0 bias)
        "test-bloom",  # MHA + ALiBi (global-head slope slicing)
    ],
)
def test_tp_block_matches_full(tmp_path, model):
    from petals_amd.models import get_model_block
    from petals_amd.models.config_base import load_model_config
    from petals_amd.server.from_pretrained import init_random_block_

    cfg = load_model_config(model)
    full = get_model_block(cfg, 0)
    init_random_block_(full, cfg, 0)
    torch.manual_seed(0)
    x = torch.randn(2, 7, cfg.hidden_size)
    ref = full(x)
    x_ref = x.detach().clone().requires_grad_(True)
    full(x_ref).square().mean().backward()

    payload_path = str(tmp_path / "payload.pt")
    result_path = str(tmp_path / "result.pt")
    torch.save({"state_dict": full.state_dict(), "x": x, "model": model}, payload_path)

    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    mp.spawn(_tp_worker, args=(2, port, payload_path, result_path), nprocs=2, join=True)

    result = torch.load(result_path, weights_only=False)
    assert torch.allclose(result["out"], ref, atol=1e-5), (result["out"] - ref).abs().max()
    assert torch.allclose(result["inc"], ref, atol=1e-4), (result["inc"] - ref).abs().max()
    assert torch.allclose(result["x_grad"], x_ref.grad, atol=1e-5), (
        (result["x_grad"] - x_ref.grad).abs().max()
    )
