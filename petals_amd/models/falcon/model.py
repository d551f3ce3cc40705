"""Distributed Falcon model classes (parity: reference models/falcon/model.py)."""

from __future__ import annotations

from torch import nn

from petals_amd.models.model_base import (
    DistributedForCausalLMBase,
    DistributedForSequenceClassificationBase,
    DistributedModelBase,
)


class DistributedFalconModel(DistributedModelBase):
    def _make_final_norm(self, config):
        return nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)


class DistributedFalconForCausalLM(DistributedForCausalLMBase):
    @classmethod
    def _build(cls, config):
        return cls(config, model=DistributedFalconModel(config))


class DistributedFalconForSequenceClassification(DistributedForSequenceClassificationBase):
    @classmethod
    def from_pretrained(cls, model_name_or_path: str, config=None, num_labels: int = 2, **kwargs):
        if config is None:
            from petals_amd.utils.auto_config import AutoDistributedConfig

            config = AutoDistributedConfig.from_pretrained(model_name_or_path, **kwargs)
        return cls(config, model=DistributedFalconModel(config), num_labels=num_labels)
