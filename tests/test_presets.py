"""Preset configs load and match the public architecture parameter counts."""

import pytest

from petals_amd.models.config_base import load_model_config


def block_params(cfg):
    h, inter, kv = cfg.hidden_size, cfg.intermediate_size, cfg.n_kv_heads * cfg.head_dim
    attn = h * h + 2 * h * kv + h * h
    n_mats = 3 if cfg.model_type in ("llama", "mixtral") else 2  # SwiGLU vs GELU MLP
    mlp = n_mats * h * inter * getattr(cfg, "num_local_experts", 1)
    return attn + mlp


@pytest.mark.parametrize("preset,total_b", [
    ("llama-2-7b", 6.5), ("llama-2-70b", 68.0), ("llama-3.1-405b", 400.0),
    ("mixtral-8x22b", 138.0), ("falcon-180b", 170.0), ("bloom-176b", 172.0),
])
def test_preset_scale(preset, total_b):
    cfg = load_model_config(preset)
    approx = block_params(cfg) * cfg.num_blocks + 2 * cfg.vocab_size * cfg.hidden_size
    assert approx / 1e9 == pytest.approx(total_b, rel=0.15), approx / 1e9


def test_all_presets_build_blocks_on_meta():
    import torch

    from petals_amd.models import get_model_block

    for preset in ("llama-2-7b", "bloom-560m", "falcon-7b", "falcon-40b", "mixtral-8x7b",
                   "llama-3.1-405b", "llama-3-8b", "bloom-176b", "falcon-180b", "mixtral-8x22b"):
        cfg = load_model_config(preset)
        with torch.device("meta"):
            blk = get_model_block(cfg, 0)
        assert sum(p.numel() for p in blk.parameters()) > 0
