"""Framework-native model configs.

We intentionally do NOT subclass transformers' config classes (the reference
does, `models/llama/config.py:16`): transformers 5.x reshuffled fields like
rope_theta -> rope_parameters, and the server must not depend on HF internals.
Configs are plain dataclasses constructed from an HF ``config.json`` dict
(both old and new field layouts accepted) or from built-in presets
(`petals_amd.models.presets`) since this environment has no network.
"""

from __future__ import annotations

import dataclasses
import json
import os
from typing import Any, ClassVar, Dict, Optional, Type

_MODEL_CONFIG_REGISTRY: Dict[str, Type["ModelConfig"]] = {}


def register_config(model_type: str):
    def deco(cls):
        _MODEL_CONFIG_REGISTRY[model_type] = cls
        cls.model_type = model_type
        return cls

    return deco


@dataclasses.dataclass
class ModelConfig:
    """Base config: fields common to every supported decoder-only family."""

    hidden_size: int = 1024
    num_hidden_layers: int = 24
    num_attention_heads: int = 16
    num_key_value_heads: Optional[int] = None
    intermediate_size: Optional[int] = None
    vocab_size: int = 32000
    max_position_embeddings: int = 2048
    layer_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    torch_dtype: str = "bfloat16"
    name_or_path: str = ""

    model_type: ClassVar[str] = ""  # set by @register_config
    # dht_prefix identifies the block namespace in the swarm: "{prefix}.{i}"
    dht_prefix: str = ""
    # block_prefix is the state-dict path of the block list, e.g. "model.layers"
    block_prefix: str = "model.layers"
    # explicit head_dim (tensor-parallel SHARD configs keep hidden_size while
    # dividing the head count, so the derived value would be wrong)
    head_dim_override: Optional[int] = None

    @property
    def head_dim(self) -> int:
        return self.head_dim_override or (self.hidden_size // self.num_attention_heads)

    @property
    def n_kv_heads(self) -> int:
        return self.num_key_value_heads or self.num_attention_heads

    @property
    def num_blocks(self) -> int:
        return self.num_hidden_layers

    def __post_init__(self):
        if self.num_key_value_heads is None:
            self.num_key_value_heads = self.num_attention_heads
        if self.intermediate_size is None:
            self.intermediate_size = 4 * self.hidden_size
        if not self.dht_prefix:
            self.dht_prefix = self.default_dht_prefix()

    def default_dht_prefix(self) -> str:
        base = self.name_or_path.split("/")[-1] if self.name_or_path else self.model_type
        base = base.replace(".", "-") or self.model_type or "model"
        return f"{base}-petals-amd"

    # ------------------------------------------------------- HF conversion

    @classmethod
    def field_names(cls):
        return {f.name for f in dataclasses.fields(cls)}

    @classmethod
    def from_hf_dict(cls, d: Dict[str, Any], name_or_path: str = "") -> "ModelConfig":
        known = cls.field_names()
        kwargs = {k: v for k, v in d.items() if k in known and k != "model_type"}
        # normalize dtype naming
        td = d.get("torch_dtype") or d.get("dtype")
        if td:
            kwargs["torch_dtype"] = str(td).replace("torch.", "")
        for alias in ("rms_norm_eps", "layer_norm_epsilon"):
            if alias in d:
                kwargs["layer_norm_eps"] = d[alias]
        if name_or_path:
            kwargs["name_or_path"] = name_or_path
        cfg = cls(**kwargs)
        cfg._absorb_hf_extras(d)
        return cfg

    def _absorb_hf_extras(self, d: Dict[str, Any]) -> None:
        """Family-specific post-processing of an HF config dict."""

    def to_dict(self) -> Dict[str, Any]:
        d = dataclasses.asdict(self)
        d["model_type"] = self.model_type
        return d


def load_model_config(path_or_preset: str) -> ModelConfig:
    """Load a config from a local HF checkpoint dir, a config.json path, or a
    built-in preset name (see petals_amd.models.presets)."""
    import petals_amd.models  # ensure registrations  # noqa: F401
    from petals_amd.models import presets

    if path_or_preset in presets.PRESETS:
        d = dict(presets.PRESETS[path_or_preset])
        name = path_or_preset
    else:
        p = path_or_preset
        if os.path.isdir(p):
            p = os.path.join(p, "config.json")
        if not os.path.exists(p):
            raise FileNotFoundError(
                f"{path_or_preset!r} is neither a preset ({sorted(presets.PRESETS)}) nor a local checkpoint"
            )
        with open(p) as f:
            d = json.load(f)
        name = path_or_preset
    model_type = d.get("model_type")
    if model_type not in _MODEL_CONFIG_REGISTRY:
        raise ValueError(f"unsupported model_type {model_type!r}; known: {sorted(_MODEL_CONFIG_REGISTRY)}")
    return _MODEL_CONFIG_REGISTRY[model_type].from_hf_dict(d, name_or_path=name)
