"""Per-block weight loading.

Parity with reference ``server/from_pretrained.py:35-224``: a server loads ONLY
the safetensors shards containing its blocks ("{block_prefix}.{i}."), never the
whole checkpoint. No-network environment: checkpoints come from local dirs;
`init_random_block` provides deterministic random weights for tests/benchmarks
(seeded by (model name, layer index) so every server materializes identical
weights for the same block without communication).
"""

from __future__ import annotations

import json
import logging
import os
from typing import Dict

import torch

from petals_amd.models import get_model_block
from petals_amd.models.config_base import ModelConfig

logger = logging.getLogger(__name__)


def _find_block_tensors(model_dir: str, prefix: str) -> Dict[str, str]:
    """Maps tensor name -> shard filename for tensors under `prefix`."""
    index_path = os.path.join(model_dir, "model.safetensors.index.json")
    if os.path.exists(index_path):
        with open(index_path) as f:
            index = json.load(f)["weight_map"]
        return {name: shard for name, shard in index.items() if name.startswith(prefix)}
    single = os.path.join(model_dir, "model.safetensors")
    if os.path.exists(single):
        from safetensors import safe_open

        with safe_open(single, framework="pt") as f:
            return {name: "model.safetensors" for name in f.keys() if name.startswith(prefix)}
    raise FileNotFoundError(f"no safetensors checkpoint found in {model_dir}")


def load_block_state_dict(model_dir: str, config: ModelConfig, block_index: int) -> Dict[str, torch.Tensor]:
    prefix = f"{config.block_prefix}.{block_index}."
    tensor_map = _find_block_tensors(model_dir, prefix)
    if not tensor_map:
        # fall back to alternate prefixes used by some checkpoints
        for alt in ("model.layers", "transformer.h", "h"):
            tensor_map = _find_block_tensors(model_dir, f"{alt}.{block_index}.")
            if tensor_map:
                prefix = f"{alt}.{block_index}."
                break
    if not tensor_map:
        raise KeyError(f"no tensors with prefix {prefix!r} in {model_dir}")
    from safetensors import safe_open

    state_dict: Dict[str, torch.Tensor] = {}
    by_shard: Dict[str, list] = {}
    for name, shard in tensor_map.items():
        by_shard.setdefault(shard, []).append(name)
    for shard, names in by_shard.items():
        with safe_open(os.path.join(model_dir, shard), framework="pt") as f:
            for name in names:
                state_dict[name[len(prefix):]] = f.get_tensor(name)
    return state_dict


def build_empty_block(config: ModelConfig, block_index: int, device, torch_dtype) -> torch.nn.Module:
    """Construct a block on the meta device (no allocation / default-init cost)
    and materialize it empty straight on the target device."""
    with torch.device("meta"):
        block = get_model_block(config, block_index)
    return block.to_empty(device=device).to(torch_dtype)


def load_pretrained_block(
    model_name_or_dir: str,
    config: ModelConfig,
    block_index: int,
    *,
    torch_dtype: torch.dtype = torch.float32,
    device: torch.device = torch.device("cpu"),
    quant_type: str = "none",
    tp_rank: int = 0,
    tp_world: int = 1,
    tp_group=None,
) -> torch.nn.Module:
    """Build one block and fill it with checkpoint weights (or deterministic
    random weights when no local checkpoint exists). With tp_world > 1 the
    result is this rank's TENSOR-PARALLEL SHARD of the block (parallel/tp.py):
    the full-block weights are materialized on CPU, sliced, and only the shard
    moves to the device."""
    device = torch.device(device)
    if tp_world > 1:
        from petals_amd.parallel.tp import build_tp_block

        full = build_empty_block(config, block_index, torch.device("cpu"), torch_dtype)
        if os.path.isdir(model_name_or_dir):
            sd = load_block_state_dict(model_name_or_dir, config, block_index)
            sd = {k: v.to(dtype=torch_dtype if v.is_floating_point() else None) for k, v in sd.items()}
            full.load_state_dict(sd, strict=False, assign=True)
        else:
            init_random_block_(full, config, block_index)
        shard = build_tp_block(config, block_index, rank=tp_rank, world=tp_world, group=tp_group)
        shard = shard.to(torch_dtype)
        shard.load_from_full_state_dict(full.state_dict())
        del full
        shard = shard.to(device).eval()
        if device.type == "cuda":
            shard.optimize_for_inference(quant=quant_type)
        return shard
    block = build_empty_block(config, block_index, device, torch_dtype)
    if os.path.isdir(model_name_or_dir):
        sd = load_block_state_dict(model_name_or_dir, config, block_index)
        sd = {k: v.to(device=device, dtype=torch_dtype if v.is_floating_point() else None) for k, v in sd.items()}
        report = block.load_state_dict(sd, strict=False, assign=True)
        if report.missing_keys:
            logger.warning("block %d: missing keys %s", block_index, report.missing_keys)
    else:
        init_random_block_(block, config, block_index)
    block = block.eval()
    if device.type == "cuda" and hasattr(block, "optimize_for_inference"):
        block.optimize_for_inference(quant=quant_type)
    return block


def init_random_block_(block: torch.nn.Module, config: ModelConfig, block_index: int) -> None:
    """Deterministic random init: same (model, block) => same weights on every
    server *of the same device type* (CPU and GPU RNG sequences differ)."""
    import zlib

    key = f"{config.name_or_path or config.model_type}:{block_index}"
    seed = (zlib.crc32(key.encode()) & 0x7FFFFFFF) or 1
    device = next(block.parameters()).device
    gen = torch.Generator(device=device).manual_seed(seed)
    with torch.no_grad():
        for name, p in sorted(block.named_parameters()):
            if p.dim() >= 2:
                p.normal_(0.0, 0.02, generator=gen)
            elif "bias" in name:
                p.zero_()
            else:  # norm weights
                p.fill_(1.0)
