"""Distributed Llama model classes (parity: reference models/llama/model.py)."""

from __future__ import annotations

import torch
from torch import nn

from petals_amd import ops
from petals_amd.models.model_base import (
    DistributedForCausalLMBase,
    DistributedForSequenceClassificationBase,
    DistributedModelBase,
)


class _RMSNormHead(nn.Module):
    def __init__(self, hidden_size: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.eps = eps

    def forward(self, x):
        return ops.rms_norm(x, self.weight, self.eps)


class DistributedLlamaModel(DistributedModelBase):
    def _make_final_norm(self, config):
        return _RMSNormHead(config.hidden_size, config.layer_norm_eps)

    @property
    def layers(self):  # HF-style alias
        return self.h


class DistributedLlamaForCausalLM(DistributedForCausalLMBase):
    @classmethod
    def _build(cls, config):
        return cls(config, model=DistributedLlamaModel(config))


class DistributedLlamaForSequenceClassification(DistributedForSequenceClassificationBase):
    @classmethod
    def from_pretrained(cls, model_name_or_path: str, config=None, num_labels: int = 2, **kwargs):
        if config is None:
            from petals_amd.utils.auto_config import AutoDistributedConfig

            config = AutoDistributedConfig.from_pretrained(model_name_or_path, **kwargs)
        return cls(config, model=DistributedLlamaModel(config), num_labels=num_labels)
