"""AutoDistributed* entry points (parity: reference utils/auto_config.py).

Maps HF `model_type` -> (DistributedConfig, DistributedModel, ...) registered
by each model family subpackage.
"""

from __future__ import annotations

import json
import os
from typing import Dict

_REGISTRY: Dict[str, Dict[str, type]] = {}


def register_model_classes(model_type: str, **classes) -> None:
    _REGISTRY[model_type] = classes


def _ensure_registered():
    import petals_amd.models as m

    m._ensure_families()
    for family in ("llama", "bloom", "falcon", "mixtral"):
        if family not in _REGISTRY:
            mod = getattr(__import__(f"petals_amd.models.{family}", fromlist=["_register_models"]), "_register_models")
            mod()


def _resolve_config_dict(name_or_path: str):
    from petals_amd.models import presets

    if name_or_path in presets.PRESETS:
        return dict(presets.PRESETS[name_or_path]), name_or_path
    p = name_or_path
    if os.path.isdir(p):
        p = os.path.join(p, "config.json")
    if not os.path.exists(p):
        raise FileNotFoundError(
            f"{name_or_path!r} is neither a preset nor a local checkpoint dir (no network in this build)"
        )
    with open(p) as f:
        return json.load(f), name_or_path


class _AutoDistributedBase:
    _kind = "config"

    @classmethod
    def from_pretrained(cls, model_name_or_path: str, **kwargs):
        _ensure_registered()
        d, name = _resolve_config_dict(model_name_or_path)
        model_type = d.get("model_type")
        if model_type not in _REGISTRY:
            raise ValueError(f"unsupported model_type {model_type!r}; known: {sorted(_REGISTRY)}")
        classes = _REGISTRY[model_type]
        config_cls = classes["config"]
        # build the *distributed* config (family config + client knobs)
        from petals_amd.models.model_base import make_distributed_config

        config = make_distributed_config(config_cls, d, name, **kwargs)
        if cls._kind == "config":
            return config
        model_cls = classes[cls._kind]
        return model_cls.from_pretrained(model_name_or_path, config=config)


class AutoDistributedConfig(_AutoDistributedBase):
    _kind = "config"


class AutoDistributedModel(_AutoDistributedBase):
    _kind = "model"


class AutoDistributedModelForCausalLM(_AutoDistributedBase):
    _kind = "model_for_causal_lm"


class AutoDistributedModelForSequenceClassification(_AutoDistributedBase):
    _kind = "model_for_sequence_classification"
