"""Distributed BLOOM model classes (parity: reference models/bloom/model.py)."""

from __future__ import annotations

import torch
from torch import nn

from petals_amd.models.model_base import (
    DistributedForCausalLMBase,
    DistributedForSequenceClassificationBase,
    DistributedModelBase,
)


class DistributedBloomModel(DistributedModelBase):
    def __init__(self, config, *, dht=None):
        super().__init__(config, dht=dht)
        self.word_embeddings_layernorm = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)

    def _make_final_norm(self, config):
        return nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)

    def _embed(self, input_ids: torch.Tensor) -> torch.Tensor:
        return self.word_embeddings_layernorm(self.embed_tokens(input_ids))


class DistributedBloomForCausalLM(DistributedForCausalLMBase):
    @classmethod
    def _build(cls, config):
        return cls(config, model=DistributedBloomModel(config))


class DistributedBloomForSequenceClassification(DistributedForSequenceClassificationBase):
    @classmethod
    def from_pretrained(cls, model_name_or_path: str, config=None, num_labels: int = 2, **kwargs):
        if config is None:
            from petals_amd.utils.auto_config import AutoDistributedConfig

            config = AutoDistributedConfig.from_pretrained(model_name_or_path, **kwargs)
        return cls(config, model=DistributedBloomModel(config), num_labels=num_labels)
