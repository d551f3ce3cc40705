"""Model family registry: configs, blocks, and distributed model classes."""

from typing import Dict, Type

import torch

from petals_amd.models.config_base import ModelConfig, load_model_config

_BLOCK_REGISTRY: Dict[str, type] = {}


def register_block(model_type: str):
    def deco(cls):
        _BLOCK_REGISTRY[model_type] = cls
        return cls

    return deco


def get_model_block(config: ModelConfig, layer_idx: int = 0) -> torch.nn.Module:
    """Block factory (parity: reference server/block_utils.py:56)."""
    _ensure_families()
    if config.model_type not in _BLOCK_REGISTRY:
        raise ValueError(f"no block class for model_type {config.model_type!r}")
    return _BLOCK_REGISTRY[config.model_type](config, layer_idx)


_loaded = False


def _ensure_families():
    global _loaded
    if not _loaded:
        import petals_amd.models.llama  # noqa: F401
        import petals_amd.models.bloom  # noqa: F401
        import petals_amd.models.falcon  # noqa: F401
        import petals_amd.models.mixtral  # noqa: F401

        _loaded = True


_ensure_families()
