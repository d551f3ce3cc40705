"""Falcon family config (parity: reference models/falcon/config.py)."""

from __future__ import annotations

import dataclasses
from typing import Any, Dict

from petals_amd.models.config_base import ModelConfig, register_config


@register_config("falcon")
@dataclasses.dataclass
class FalconConfig(ModelConfig):
    num_kv_heads: int = 1
    multi_query: bool = True
    parallel_attn: bool = True
    new_decoder_architecture: bool = False
    bias: bool = False
    rope_theta: float = 10000.0
    block_prefix: str = "transformer.h"

    def __post_init__(self):
        if self.intermediate_size is None:
            self.intermediate_size = 4 * self.hidden_size
        super().__post_init__()
        # falcon's n_kv semantics: new arch uses num_kv_heads, old uses multi_query
        if self.new_decoder_architecture:
            self.num_key_value_heads = self.num_kv_heads
        elif self.multi_query:
            self.num_key_value_heads = 1
        else:
            self.num_key_value_heads = self.num_attention_heads

    def default_dht_prefix(self) -> str:
        base = self.name_or_path.split("/")[-1] if self.name_or_path else "falcon"
        return f"{base.replace('.', '-')}-petals-amd"

    def _absorb_hf_extras(self, d: Dict[str, Any]) -> None:
        rp = d.get("rope_parameters")
        if isinstance(rp, dict):
            self.rope_theta = float(rp.get("rope_theta", self.rope_theta))
        if "rope_theta" in d:
            self.rope_theta = float(d["rope_theta"])
