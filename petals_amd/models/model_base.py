"""Shared scaffolding for Distributed* model classes.

Every family provides:
  DistributedXModel        — local embeddings + RemoteSequential + final norm
  DistributedXForCausalLM  — + client-local LM head + generation mixin
  DistributedXForSequenceClassification — + score head

The distributed config is the family ModelConfig with ClientConfig / PTune /
LMHead knobs attached (parity: reference models/*/config.py multiple
inheritance of HF config + ClientConfig + PTuneConfig + LMHeadConfig).
"""

from __future__ import annotations

import dataclasses
import json
import logging
import os
from typing import Optional, Tuple

import torch
from torch import nn

from petals_amd.client.config import ClientConfig
from petals_amd.client.lm_head import LMHead, LMHeadConfig
from petals_amd.client.ptune import PTuneConfig, PTuneMixin
from petals_amd.client.remote_generation import RemoteGenerationMixin
from petals_amd.client.remote_sequential import RemoteSequential
from petals_amd.utils.misc import DUMMY

logger = logging.getLogger(__name__)

_CLIENT_FIELD_NAMES = {f.name for f in dataclasses.fields(ClientConfig)}
_PTUNE_FIELD_NAMES = {f.name for f in dataclasses.fields(PTuneConfig)}
_LMHEAD_FIELD_NAMES = {f.name for f in dataclasses.fields(LMHeadConfig)}


def make_distributed_config(config_cls, hf_dict: dict, name: str, **kwargs):
    """Build the family config from an HF dict, then attach client knobs from
    kwargs (initial_peers=..., pre_seq_len=..., tuning_mode=..., etc.)."""
    config = config_cls.from_hf_dict(hf_dict, name_or_path=name)
    client_kwargs = {k: v for k, v in kwargs.items() if k in _CLIENT_FIELD_NAMES}
    ptune_kwargs = {k: v for k, v in kwargs.items() if k in _PTUNE_FIELD_NAMES}
    lmhead_kwargs = {k: v for k, v in kwargs.items() if k in _LMHEAD_FIELD_NAMES}
    known = _CLIENT_FIELD_NAMES | _PTUNE_FIELD_NAMES | _LMHEAD_FIELD_NAMES | {"dht", "torch_dtype"}
    unknown = set(kwargs) - known - config_cls.field_names()
    if unknown:
        raise TypeError(f"unknown config kwargs: {sorted(unknown)}")
    config.client = ClientConfig(**client_kwargs)
    if config.client.dht_prefix:
        config.dht_prefix = config.client.dht_prefix
    ptune = PTuneConfig(**ptune_kwargs)
    config.pre_seq_len = ptune.pre_seq_len
    config.tuning_mode = ptune.tuning_mode
    lmh = LMHeadConfig(**lmhead_kwargs)
    config.use_chunked_forward = lmh.use_chunked_forward
    config.chunked_forward_step = lmh.chunked_forward_step
    for k, v in kwargs.items():
        if k in config_cls.field_names():
            setattr(config, k, v)
    return config


@dataclasses.dataclass
class ModelOutput:
    logits: Optional[torch.Tensor] = None
    last_hidden_state: Optional[torch.Tensor] = None
    hidden_states: Optional[Tuple] = None


class DistributedModelBase(nn.Module, PTuneMixin):
    """Embeddings (local) -> RemoteSequential -> final norm (local)."""

    def __init__(self, config, *, dht=None):
        super().__init__()
        self.config = config
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.h = RemoteSequential(config, dht=dht)
        self.norm = self._make_final_norm(config)
        self.init_prompts(config)
        self.requires_grad_(False)
        if config.tuning_mode and "ptune" in config.tuning_mode:
            self.prompt_embeddings.requires_grad_(True)
            if config.tuning_mode == "deep_ptune":
                self.intermediate_prompt_embeddings.requires_grad_(True)

    # subclasses override
    def _make_final_norm(self, config) -> nn.Module:
        raise NotImplementedError

    def _embed(self, input_ids: torch.Tensor) -> torch.Tensor:
        return self.embed_tokens(input_ids)

    def _prepare_embeds(self, inputs_embeds: torch.Tensor) -> torch.Tensor:
        """Applied to the embedding stream regardless of whether it came from
        input_ids or caller-provided inputs_embeds (BLOOM overrides with its
        embedding LayerNorm)."""
        return inputs_embeds

    @property
    def word_embeddings(self):  # bloom-style alias
        return self.embed_tokens

    @classmethod
    def from_pretrained(cls, model_name_or_path: str, config=None, torch_dtype=torch.float32, **kwargs):
        if config is None:
            from petals_amd.utils.auto_config import AutoDistributedConfig

            config = AutoDistributedConfig.from_pretrained(model_name_or_path, **kwargs)
        model = cls(config)
        if os.path.isdir(model_name_or_path):
            _load_client_side_weights(model, model_name_or_path, config)
        else:
            _init_client_side_weights(model, config)
        return model.to(torch_dtype)

    def forward(
        self,
        input_ids: Optional[torch.Tensor] = None,
        inputs_embeds: Optional[torch.Tensor] = None,
        hypo_ids: Optional[torch.Tensor] = None,
        **kwargs,
    ) -> ModelOutput:
        assert (input_ids is None) != (inputs_embeds is None), "provide input_ids xor inputs_embeds"
        if inputs_embeds is None:
            input_ids = input_ids.to(self.embed_tokens.weight.device)
            inputs_embeds = self._embed(input_ids)

        batch = inputs_embeds.shape[0]
        use_prompts = self.tuning_mode and "ptune" in self.tuning_mode
        session = self.h.active_session
        at_start = session is None or session.position == 0
        intermediate_prompts = DUMMY
        if use_prompts and at_start:
            prompts, intermediate_prompts = self.get_prompt(batch)
            inputs_embeds = torch.cat([prompts.to(inputs_embeds.dtype), inputs_embeds], dim=1)
        # family hook applied to BOTH input_ids and caller-provided
        # inputs_embeds, after prompt concat (BLOOM's embedding LayerNorm —
        # reference models/bloom/model.py:83 applies it exactly here)
        inputs_embeds = self._prepare_embeds(inputs_embeds)

        step_kwargs = {}
        if session is not None and hypo_ids is not None:
            step_kwargs["hypo_ids"] = hypo_ids

        hidden = self.h(inputs_embeds, prompts=intermediate_prompts, **step_kwargs)

        if use_prompts and at_start:
            hidden = hidden[:, self.pre_seq_len :]
        hidden = self.norm(hidden)
        return ModelOutput(last_hidden_state=hidden)


class DistributedForCausalLMBase(nn.Module, RemoteGenerationMixin):
    model_attr = "transformer"

    def __init__(self, config, *, dht=None, model: Optional[DistributedModelBase] = None):
        super().__init__()
        self.config = config
        self.transformer = model
        self.lm_head = LMHead(config)
        self._next_hypo_ids: Optional[torch.Tensor] = None

    # convenience aliases matching each family's HF naming
    @property
    def model(self):
        return self.transformer

    @property
    def pre_seq_len(self):
        return getattr(self.transformer, "pre_seq_len", 0)

    @property
    def tuning_mode(self):
        return getattr(self.transformer, "tuning_mode", None)

    def forward(self, input_ids: Optional[torch.Tensor] = None, labels=None, **kwargs) -> ModelOutput:
        hypo_ids = self._next_hypo_ids
        self._next_hypo_ids = None
        out = self.transformer(input_ids=input_ids, hypo_ids=hypo_ids, **kwargs)
        logits = self.lm_head(out.last_hidden_state)
        return ModelOutput(logits=logits, last_hidden_state=out.last_hidden_state)

    def get_input_embeddings(self):
        return self.transformer.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    # --------------------------------------------------------- checkpoint IO

    @classmethod
    def from_pretrained(cls, model_name_or_path: str, config=None, torch_dtype=torch.float32, **kwargs):
        if config is None:
            from petals_amd.utils.auto_config import AutoDistributedConfig

            config = AutoDistributedConfig.from_pretrained(model_name_or_path, **kwargs)
        model = cls._build(config)
        if os.path.isdir(model_name_or_path):
            _load_client_side_weights(model, model_name_or_path, config)
        else:
            _init_client_side_weights(model, config)
        return model.to(torch_dtype)

    @classmethod
    def _build(cls, config):
        raise NotImplementedError


class DistributedForSequenceClassificationBase(nn.Module):
    def __init__(self, config, *, model: DistributedModelBase, num_labels: int = 2):
        super().__init__()
        self.config = config
        self.num_labels = num_labels
        self.transformer = model
        self.score = nn.Linear(config.hidden_size, num_labels, bias=False)

    def forward(self, input_ids: Optional[torch.Tensor] = None, **kwargs) -> ModelOutput:
        out = self.transformer(input_ids=input_ids, **kwargs)
        logits = self.score(out.last_hidden_state[:, -1])
        return ModelOutput(logits=logits, last_hidden_state=out.last_hidden_state)


# ------------------------------------------------------------------ weights


def _client_side_names(config):
    """(embeddings name, final norm name, lm head name) in HF state dicts."""
    mt = config.model_type
    if mt in ("llama", "mixtral"):
        return "model.embed_tokens.weight", "model.norm.weight", "lm_head.weight"
    if mt == "bloom":
        return "transformer.word_embeddings.weight", "transformer.ln_f.weight", "lm_head.weight"
    if mt == "falcon":
        return "transformer.word_embeddings.weight", "transformer.ln_f.weight", "lm_head.weight"
    raise ValueError(mt)


def _load_client_side_weights(model, model_dir: str, config):
    """Load ONLY embeddings/norm/head shards (parity: client/from_pretrained.py —
    shards for block layers are skipped)."""
    from safetensors import safe_open

    emb_name, norm_name, head_name = _client_side_names(config)
    wanted = {emb_name, norm_name, head_name}
    if config.model_type == "bloom":
        wanted |= {"transformer.ln_f.bias", "transformer.word_embeddings_layernorm.weight",
                   "transformer.word_embeddings_layernorm.bias"}
    index_path = os.path.join(model_dir, "model.safetensors.index.json")
    files = set()
    if os.path.exists(index_path):
        with open(index_path) as f:
            weight_map = json.load(f)["weight_map"]
        for name in wanted:
            if name in weight_map:
                files.add(weight_map[name])
    else:
        files = {"model.safetensors"}
    found = {}
    for fname in files:
        with safe_open(os.path.join(model_dir, fname), framework="pt") as f:
            for name in wanted:
                if name in f.keys():
                    found[name] = f.get_tensor(name)
    tfm = getattr(model, "transformer", None) or model
    if emb_name in found:
        tfm.embed_tokens.weight.data = found[emb_name].to(tfm.embed_tokens.weight.dtype)
    if norm_name in found:
        tfm.norm.weight.data = found[norm_name].to(tfm.norm.weight.dtype)
        if hasattr(tfm.norm, "bias") and "transformer.ln_f.bias" in found:
            tfm.norm.bias.data = found["transformer.ln_f.bias"]
    if hasattr(model, "lm_head"):
        if head_name in found:
            model.lm_head.weight.data = found[head_name]
        elif config.tie_word_embeddings or head_name not in found:
            model.lm_head.weight.data = tfm.embed_tokens.weight.data
    if config.model_type == "bloom" and hasattr(tfm, "word_embeddings_layernorm"):
        w = found.get("transformer.word_embeddings_layernorm.weight")
        b = found.get("transformer.word_embeddings_layernorm.bias")
        if w is not None:
            tfm.word_embeddings_layernorm.weight.data = w
        if b is not None:
            tfm.word_embeddings_layernorm.bias.data = b


def _init_client_side_weights(model, config):
    """Deterministic random init consistent with server-side random blocks."""
    import zlib

    key = f"{config.name_or_path or config.model_type}:client"
    seed = (zlib.crc32(key.encode()) & 0x7FFFFFFF) or 1
    gen = torch.Generator().manual_seed(seed)
    tfm = getattr(model, "transformer", None) or model
    with torch.no_grad():
        tfm.embed_tokens.weight.normal_(0, 0.02, generator=gen)
        if hasattr(model, "lm_head"):
            model.lm_head.weight.data = tfm.embed_tokens.weight.data
