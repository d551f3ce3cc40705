"""Server self-benchmark: how many tokens/sec can this device push through one
block? Feeds routing (ServerInfo.throughput) and block-span sizing.

Parity with reference ``server/throughput.py:37-255``: measures inference RPS
(1 token x N steps) and forward RPS (1024 tokens x K steps) per block, caches
results to a JSON file keyed by (device, dtype, quant, model shape). Network
RPS uses a configurable bandwidth assumption (no speedtest in this offline
environment; xGMI/TCP peers are much faster than the reference's 100 Mbit
default anyway).
"""

from __future__ import annotations

import fcntl
import json
import logging
import os
import time
from pathlib import Path
from typing import Dict, Optional

import torch

from petals_amd.models.config_base import ModelConfig
from petals_amd.server.from_pretrained import init_random_block_

logger = logging.getLogger(__name__)

DEFAULT_CACHE_PATH = Path(os.environ.get("PETALS_AMD_CACHE", Path.home() / ".cache" / "petals_amd"))
THROUGHPUT_FILE = "throughput_v1.json"


def get_server_throughput(
    config: ModelConfig,
    *,
    device: torch.device,
    dtype: torch.dtype,
    num_blocks: int = 1,
    network_gbps: float = 10.0,
    quant_type: str = "none",
    force_eval: bool = False,
    cache_dir: Optional[Path] = None,
) -> Dict[str, float]:
    """Returns {"throughput": min(compute/blocks, network), "inference_rps": ...,
    "forward_rps": ..., "network_rps": ...}."""
    cache_dir = Path(cache_dir or DEFAULT_CACHE_PATH)
    cache_dir.mkdir(parents=True, exist_ok=True)
    cache_path = cache_dir / THROUGHPUT_FILE
    key = json.dumps(
        {
            "model": config.name_or_path or config.model_type,
            "hidden": config.hidden_size,
            "device": str(device),
            "dtype": str(dtype),
            "quant": quant_type,
        },
        sort_keys=True,
    )

    cache = {}
    if cache_path.exists():
        try:
            with open(cache_path) as f:
                fcntl.flock(f, fcntl.LOCK_SH)
                cache = json.load(f)
        except (json.JSONDecodeError, OSError):
            cache = {}
    if not force_eval and key in cache:
        entry = cache[key]
    else:
        entry = measure_compute_rps(config, device=device, dtype=dtype)
        cache[key] = entry
        try:
            with open(cache_path, "w") as f:
                fcntl.flock(f, fcntl.LOCK_EX)
                json.dump(cache, f)
        except OSError as e:
            logger.warning("could not write throughput cache: %r", e)

    network_rps = network_gbps * 1e9 / 8 / (config.hidden_size * 2)  # bf16 activations
    throughput = min(entry["forward_rps"] / max(1, num_blocks), network_rps)
    return {
        "throughput": throughput,
        "inference_rps": entry["inference_rps"],
        "forward_rps": entry["forward_rps"],
        "network_rps": network_rps,
    }


@torch.inference_mode()
def measure_compute_rps(
    config: ModelConfig,
    *,
    device: torch.device,
    dtype: torch.dtype,
    n_infer_steps: int = 20,
    n_forward_steps: int = 5,
    forward_tokens: int = 1024,
) -> Dict[str, float]:
    device = torch.device(device)
    from petals_amd.server.from_pretrained import build_empty_block, init_random_block_

    block = build_empty_block(config, 0, device, dtype)
    init_random_block_(block, config, 0)
    block = block.eval()

    k_shape, v_shape = block.kv_cache_shape(1, 1024)
    k = torch.zeros(k_shape, device=device, dtype=dtype)
    v = torch.zeros(v_shape, device=device, dtype=dtype)
    x1 = torch.randn(1, 1, config.hidden_size, device=device, dtype=dtype)

    block(x1, kv_cache=(k, v), prefix_length=0)  # warmup
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    t0 = time.perf_counter()
    for i in range(n_infer_steps):
        block(x1, kv_cache=(k, v), prefix_length=i + 1)
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    inference_rps = n_infer_steps / (time.perf_counter() - t0)

    xf = torch.randn(1, forward_tokens, config.hidden_size, device=device, dtype=dtype)
    block(xf)
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    t0 = time.perf_counter()
    for _ in range(n_forward_steps):
        block(xf)
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    forward_rps = n_forward_steps * forward_tokens / (time.perf_counter() - t0)

    del block, k, v
    if device.type == "cuda":
        torch.cuda.empty_cache()
    return {"inference_rps": inference_rps, "forward_rps": forward_rps}
