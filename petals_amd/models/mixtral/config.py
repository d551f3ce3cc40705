"""Mixtral (SMoE) config (parity: reference models/mixtral/config.py)."""

from __future__ import annotations

import dataclasses
from typing import Any, Dict, Optional

from petals_amd.models.config_base import ModelConfig, register_config


@register_config("mixtral")
@dataclasses.dataclass
class MixtralConfig(ModelConfig):
    rope_theta: float = 1000000.0
    rope_scaling: Optional[Dict[str, Any]] = None
    attention_bias: bool = False
    num_local_experts: int = 8
    num_experts_per_tok: int = 2
    block_prefix: str = "model.layers"

    def default_dht_prefix(self) -> str:
        base = self.name_or_path.split("/")[-1] if self.name_or_path else "mixtral"
        return f"{base.replace('.', '-')}-petals-amd"

    def _absorb_hf_extras(self, d: Dict[str, Any]) -> None:
        rp = d.get("rope_parameters")
        if isinstance(rp, dict):
            self.rope_theta = float(rp.get("rope_theta", self.rope_theta))
        if "rope_theta" in d:
            self.rope_theta = float(d["rope_theta"])
