"""Llama family config (Llama 1/2/3, TinyLlama, CodeLlama...).

Parity target: reference `models/llama/config.py:16-47` (DistributedLlamaConfig
with dht_prefix derivation) — ours is a native dataclass, see config_base.py.
"""

from __future__ import annotations

import dataclasses
from typing import Any, Dict, Optional

from petals_amd.models.config_base import ModelConfig, register_config


@register_config("llama")
@dataclasses.dataclass
class LlamaConfig(ModelConfig):
    rope_theta: float = 10000.0
    rope_scaling: Optional[Dict[str, Any]] = None
    attention_bias: bool = False
    mlp_bias: bool = False
    layer_norm_eps: float = 1e-5  # rms_norm_eps in HF terms
    block_prefix: str = "model.layers"

    def default_dht_prefix(self) -> str:
        base = self.name_or_path.split("/")[-1] if self.name_or_path else "llama"
        base = base.replace(".", "-")
        # reference uses "-hf" stripping; keep simple, stable naming
        return f"{base}-petals-amd"

    def _absorb_hf_extras(self, d: Dict[str, Any]) -> None:
        rp = d.get("rope_parameters")
        if isinstance(rp, dict):  # transformers >= 5 layout
            self.rope_theta = float(rp.get("rope_theta", self.rope_theta))
            rtype = rp.get("rope_type", "default")
            if rtype not in ("default", None):
                self.rope_scaling = {k: v for k, v in rp.items() if k != "rope_theta"}
        if d.get("rope_scaling"):
            self.rope_scaling = d["rope_scaling"]
