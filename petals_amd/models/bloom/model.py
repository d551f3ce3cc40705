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

    def _prepare_embeds(self, inputs_embeds: torch.Tensor) -> torch.Tensor:
        # applied after prompt concat, to ids-derived AND caller-provided
        # embeddings alike (parity: reference models/bloom/model.py:83; HF
        # BloomModel layer-norms inputs_embeds in both paths)
        return self.word_embeddings_layernorm(inputs_embeds)


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
