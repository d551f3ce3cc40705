"""CPU tests for auxiliary subsystems: disk LRU cache, throughput self-bench
cache (fcntl-locked JSON), ping EMA, constants (SURVEY §5 parity rows)."""

import math
import time

import torch

from petals_amd.models.config_base import load_model_config
from petals_amd.utils.disk_cache import allow_cache_reads, allow_cache_writes, free_disk_space_for


def test_disk_cache_lru_eviction(tmp_path):
    # three entries with distinct access times, 1 KiB each
    for i, name in enumerate(("old", "mid", "new")):
        d = tmp_path / name
        d.mkdir()
        (d / "blob").write_bytes(b"x" * 1024)
        atime = time.time() - (3 - i) * 1000
        import os

        os.utime(d, (atime, atime))
        os.utime(d / "blob", (atime, atime))

    with allow_cache_reads(tmp_path):
        pass  # shared lock acquires and releases

    # budget for ~2 entries: the least-recently-used one must go
    with allow_cache_writes(tmp_path):
        free_disk_space_for(0, cache_dir=tmp_path, max_disk_space=2 * 1024 + 512)
    left = sorted(p.name for p in tmp_path.iterdir() if p.name != "blocks.lock")
    assert "old" not in left, left
    assert "new" in left, left


def test_throughput_cache_roundtrip(tmp_path):
    from petals_amd.server.throughput import get_server_throughput

    cfg = load_model_config("test-llama")
    t0 = time.perf_counter()
    r1 = get_server_throughput(
        cfg, device=torch.device("cpu"), dtype=torch.float32, num_blocks=2,
        network_gbps=1.0, cache_dir=tmp_path,
    )
    first = time.perf_counter() - t0
    assert r1["inference_rps"] > 0 and r1["forward_rps"] > 0
    # network term: 1 Gbps / (hidden * 2 bytes)
    assert math.isclose(r1["network_rps"], 1e9 / 8 / (cfg.hidden_size * 2), rel_tol=1e-6)
    assert r1["throughput"] <= r1["network_rps"] + 1e-6

    t0 = time.perf_counter()
    r2 = get_server_throughput(
        cfg, device=torch.device("cpu"), dtype=torch.float32, num_blocks=2,
        network_gbps=1.0, cache_dir=tmp_path,
    )
    second = time.perf_counter() - t0
    assert r2["inference_rps"] == r1["inference_rps"]  # served from the JSON cache
    assert second < first  # no re-measurement


def test_ping_ema_and_expiry():
    from petals_amd.utils.ping import PingAggregator

    agg = PingAggregator(p2p=None, ema_alpha=0.5, expiration=0.2)
    agg.ping_emas["peer"] = (0.1, time.monotonic())
    d = agg.to_dict()
    assert math.isclose(d["peer"], 0.1)
    agg.ping_emas["stale"] = (0.2, time.monotonic() - 1.0)
    assert "stale" not in agg.to_dict()


def test_constants_and_misc():
    from petals_amd.constants import DEFAULT_DHT_PORT, DTYPE_MAP, PUBLIC_INITIAL_PEERS
    from petals_amd.utils.misc import sample_up_to

    assert DTYPE_MAP["bfloat16"] is torch.bfloat16 and DTYPE_MAP["auto"] is None
    assert isinstance(DEFAULT_DHT_PORT, int) and PUBLIC_INITIAL_PEERS == []
    picked = sample_up_to(range(10), 3)
    assert len(picked) == 3 and set(picked) <= set(range(10))
    assert sample_up_to([1, 2], 5) == [1, 2]
