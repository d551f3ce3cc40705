"""BLOOM family config (parity: reference models/bloom/config.py)."""

from __future__ import annotations

import dataclasses
from typing import Any, Dict, Optional

from petals_amd.models.config_base import ModelConfig, register_config


@register_config("bloom")
@dataclasses.dataclass
class BloomConfig(ModelConfig):
    apply_residual_connection_post_layernorm: bool = False
    block_prefix: str = "h"  # state dict path: transformer.h.{i}
    # TP shard configs: ALiBi slopes depend on the GLOBAL head index, so a
    # shard carries its slice window over the full model's slope table
    alibi_start_head: int = 0
    alibi_total_heads: Optional[int] = None

    def __post_init__(self):
        if self.intermediate_size is None:
            self.intermediate_size = 4 * self.hidden_size
        super().__post_init__()

    def default_dht_prefix(self) -> str:
        base = self.name_or_path.split("/")[-1] if self.name_or_path else "bloom"
        return f"{base.replace('.', '-')}-petals-amd"

    def _absorb_hf_extras(self, d: Dict[str, Any]) -> None:
        if "n_head" in d:
            self.num_attention_heads = d["n_head"]
            self.num_key_value_heads = d["n_head"]
        if "n_layer" in d:
            self.num_hidden_layers = d["n_layer"]
        if "n_embed" in d or "hidden_size" in d:
            self.hidden_size = d.get("hidden_size", d.get("n_embed"))
