"""Distributed Mixtral model classes (parity: reference models/mixtral/model.py)."""

from __future__ import annotations

from petals_amd.models.llama.model import _RMSNormHead
from petals_amd.models.model_base import (
    DistributedForCausalLMBase,
    DistributedForSequenceClassificationBase,
    DistributedModelBase,
)


class DistributedMixtralModel(DistributedModelBase):
    def _make_final_norm(self, config):
        return _RMSNormHead(config.hidden_size, config.layer_norm_eps)

    @property
    def layers(self):
        return self.h


class DistributedMixtralForCausalLM(DistributedForCausalLMBase):
    @classmethod
    def _build(cls, config):
        return cls(config, model=DistributedMixtralModel(config))


class DistributedMixtralForSequenceClassification(DistributedForSequenceClassificationBase):
    @classmethod
    def from_pretrained(cls, model_name_or_path: str, config=None, num_labels: int = 2, **kwargs):
        if config is None:
            from petals_amd.utils.auto_config import AutoDistributedConfig

            config = AutoDistributedConfig.from_pretrained(model_name_or_path, **kwargs)
        return cls(config, model=DistributedMixtralModel(config), num_labels=num_labels)
