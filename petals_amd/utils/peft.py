"""LoRA adapter serving.

Parity with reference ``utils/peft.py`` (load_peft :72, AdapterContextMixin
:138, add_adapter_to_block :212): a server loads LoRA adapters at startup;
every request selects one by name ("active_adapter" metadata) through a
context manager; blocks add the low-rank delta  y += scale * (x @ A^T) @ B^T
around each projection.

Adapters load from local dirs in the standard PEFT layout
(adapter_config.json + adapter_model.safetensors with keys like
``base_model.model.model.layers.{i}.self_attn.q_proj.lora_A.weight``).
"""

from __future__ import annotations

import contextlib
import contextvars
import dataclasses
import json
import os
from typing import Dict, Optional, Tuple

import torch

_active_adapter: contextvars.ContextVar[Optional[str]] = contextvars.ContextVar(
    "petals_amd_active_adapter", default=None
)

# projection keys supported per block (llama/mixtral naming)
PROJ_KEYS = ("q", "k", "v", "o", "gate", "up", "down", "qkv", "dense", "h4h", "4hh")
_PROJ_PATHS = {
    # llama family
    "q": "self_attn.q_proj",
    "k": "self_attn.k_proj",
    "v": "self_attn.v_proj",
    "o": "self_attn.o_proj",
    "gate": "mlp.gate_proj",
    "up": "mlp.up_proj",
    "down": "mlp.down_proj",
    # falcon / bloom family (fused-QKV checkpoints)
    "qkv": "self_attention.query_key_value",
    "dense": "self_attention.dense",
    "h4h": "mlp.dense_h_to_4h",
    "4hh": "mlp.dense_4h_to_h",
}


@contextlib.contextmanager
def using_adapter(name: Optional[str]):
    token = _active_adapter.set(name)
    try:
        yield
    finally:
        _active_adapter.reset(token)


def get_active_adapter() -> Optional[str]:
    return _active_adapter.get()


@dataclasses.dataclass
class BlockAdapter:
    """Per-block LoRA weights: proj key -> (A [r, in], B [out, r], scale)."""

    name: str
    projections: Dict[str, Tuple[torch.Tensor, torch.Tensor, float]]

    def to(self, device, dtype):
        self.projections = {
            k: (a.to(device=device, dtype=dtype), b.to(device=device, dtype=dtype), s)
            for k, (a, b, s) in self.projections.items()
        }
        return self

    def delta(self, key: str, x: torch.Tensor) -> Optional[torch.Tensor]:
        entry = self.projections.get(key)
        if entry is None:
            return None
        a, b, scale = entry
        xa = x.to(a.dtype) @ a.t()
        return (xa @ b.t()).to(x.dtype) * scale

    def memory_bytes(self) -> int:
        return sum(a.numel() * a.element_size() + b.numel() * b.element_size()
                   for a, b, _ in self.projections.values())


def load_block_adapter(adapter_dir: str, block_index: int, block_prefix: str = "model.layers") -> Optional[BlockAdapter]:
    """Extract one block's LoRA tensors from a local PEFT checkpoint dir."""
    from safetensors import safe_open

    cfg_path = os.path.join(adapter_dir, "adapter_config.json")
    scaling = 1.0
    if os.path.exists(cfg_path):
        with open(cfg_path) as f:
            cfg = json.load(f)
        r = cfg.get("r", 8)
        alpha = cfg.get("lora_alpha", r)
        scaling = alpha / max(r, 1)
    weights_path = os.path.join(adapter_dir, "adapter_model.safetensors")
    if not os.path.exists(weights_path):
        raise FileNotFoundError(f"no adapter_model.safetensors in {adapter_dir}")

    projections: Dict[str, Tuple[torch.Tensor, torch.Tensor, float]] = {}
    with safe_open(weights_path, framework="pt") as f:
        keys = list(f.keys())
        for proj, path in _PROJ_PATHS.items():
            a_key = next(
                (k for k in keys if f"{block_prefix}.{block_index}.{path}.lora_A" in k), None
            )
            b_key = next(
                (k for k in keys if f"{block_prefix}.{block_index}.{path}.lora_B" in k), None
            )
            if a_key and b_key:
                projections[proj] = (f.get_tensor(a_key), f.get_tensor(b_key), scaling)
    if not projections:
        return None
    name = os.path.basename(os.path.normpath(adapter_dir))
    return BlockAdapter(name=name, projections=projections)


def add_adapter_to_block(block: torch.nn.Module, adapter: BlockAdapter) -> None:
    if not hasattr(block, "_adapters"):
        block._adapters = {}
    p = next(block.parameters(), None)
    if p is not None and p.numel():
        adapter.to(p.device, p.dtype if p.dtype.is_floating_point else torch.float32)
    block._adapters[adapter.name] = adapter
    fast = getattr(block, "_fast", None)
    if fast is not None:
        adapter.to(fast.device, torch.bfloat16)


def active_block_adapter(block: torch.nn.Module) -> Optional[BlockAdapter]:
    name = get_active_adapter()
    if not name:
        return None
    adapters = getattr(block, "_adapters", None)
    if adapters is None:
        raise KeyError(f"server has no adapters loaded but {name!r} was requested")
    if name not in adapters:
        raise KeyError(f"unknown adapter {name!r}; loaded: {sorted(adapters)}")
    return adapters[name]


def estimate_adapter_memory_per_block(adapter_dir: str, block_index: int = 0) -> int:
    ad = load_block_adapter(adapter_dir, block_index)
    return ad.memory_bytes() if ad else 0
