"""Native Mixtral decoder block: Llama-style GQA attention + sparse MoE FFN
(8 experts, top-2 router, softmax-after-topk normalization as in HF Mixtral).

Weight names match HF Mixtral per-layer state dicts:
  block_sparse_moe.gate.weight, block_sparse_moe.experts.{i}.{w1,w2,w3}.weight
(contrast: reference models/mixtral/block.py wraps HF MixtralDecoderLayer; the
experts there run densely inside HF code. Here routing is explicit so the MoE
grouped-GEMM HIP kernel can slot in on GPU.)
"""

from __future__ import annotations

from typing import Tuple

import torch
import torch.nn.functional as F
from torch import nn

from petals_amd import ops
from petals_amd.models.llama.block import LlamaAttention, RMSNorm
from petals_amd.models.mixtral.config import MixtralConfig


class MixtralExpert(nn.Module):
    def __init__(self, config: MixtralConfig):
        super().__init__()
        self.w1 = nn.Linear(config.hidden_size, config.intermediate_size, bias=False)  # gate
        self.w2 = nn.Linear(config.intermediate_size, config.hidden_size, bias=False)  # down
        self.w3 = nn.Linear(config.hidden_size, config.intermediate_size, bias=False)  # up

    def forward(self, x):
        return self.w2(ops.swiglu(self.w1(x), self.w3(x)))


class MixtralSparseMoeBlock(nn.Module):
    def __init__(self, config: MixtralConfig):
        super().__init__()
        self.num_experts = config.num_local_experts
        self.top_k = config.num_experts_per_tok
        self.gate = nn.Linear(config.hidden_size, self.num_experts, bias=False)
        self.experts = nn.ModuleList(MixtralExpert(config) for _ in range(self.num_experts))

    def forward(self, hidden_states: torch.Tensor) -> torch.Tensor:
        b, s, h = hidden_states.shape
        x = hidden_states.reshape(-1, h)
        router_logits = self.gate(x)
        probs = F.softmax(router_logits, dim=-1, dtype=torch.float32)
        weights, selected = torch.topk(probs, self.top_k, dim=-1)
        weights = weights / weights.sum(dim=-1, keepdim=True)
        weights = weights.to(x.dtype)

        out = torch.zeros_like(x)
        expert_mask = F.one_hot(selected, num_classes=self.num_experts).permute(2, 1, 0)
        for e in range(self.num_experts):
            idx, top_x = torch.where(expert_mask[e])
            if top_x.numel() == 0:
                continue
            current = x[top_x]
            out.index_add_(0, top_x, self.experts[e](current) * weights[top_x, idx, None])
        return out.reshape(b, s, h)


class MixtralBlock(nn.Module):
    _fast = None

    def optimize_for_inference(self, quant: str = "none") -> "MixtralBlock":
        """MI355X fast path: fused attention decode + routed expert GEMVs."""
        from petals_amd import ops as _ops
        from petals_amd.ops.fused_moe import MixtralFastPath

        hip = _ops._load_hip_ops()
        if hip is None:
            raise RuntimeError(
                f"cannot optimize block for MI355X: HIP extension missing ({_ops._hip_import_error!r})"
            )
        gq = self.config.num_attention_heads // self.config.n_kv_heads
        if self.config.head_dim not in (64, 128) or gq not in (1, 2, 4, 6, 8, 16):
            import logging

            logging.getLogger(__name__).warning(
                "mixtral block geometry outside the fused fast path; serving via generic ops"
            )
            return self
        self._fast = MixtralFastPath(self, hip, quant=quant)
        return self

    def __init__(self, config: MixtralConfig, layer_idx: int = 0):
        super().__init__()
        self.config = config
        self.layer_idx = layer_idx
        self.self_attn = LlamaAttention(config)
        self.block_sparse_moe = MixtralSparseMoeBlock(config)
        self.input_layernorm = RMSNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.post_attention_layernorm = RMSNorm(config.hidden_size, eps=config.layer_norm_eps)

    def forward(self, hidden_states, kv_cache=None, prefix_length: int = 0, ctx=None):
        if self._fast is not None:
            if torch.is_grad_enabled() and hidden_states.requires_grad:
                assert kv_cache is None
                return self._fast.forward_autograd(hidden_states, prefix_length)
            if kv_cache is not None and hidden_states.shape[1] == 1 and hidden_states.shape[0] <= 8:
                from petals_amd.ops.fused_decode import decode_step_auto

                # moe gemv handles B*top_k <= 16 routed rows per launch; the
                # dense-projection gemvs handle batch 8 natively
                return decode_step_auto(
                    self._fast, hidden_states, kv_cache[0], kv_cache[1], prefix_length, ctx=ctx,
                    max_b=min(8, 16 // max(self._fast.top_k, 1)),
                )
            return self._fast.forward(hidden_states, kv_cache, prefix_length)

        residual = hidden_states
        hidden_states = self.input_layernorm(hidden_states)
        hidden_states = self.self_attn(hidden_states, kv_cache=kv_cache, prefix_length=prefix_length)
        hidden_states = residual + hidden_states

        residual = hidden_states
        hidden_states = self.post_attention_layernorm(hidden_states)
        return residual + self.block_sparse_moe(hidden_states)

    def kv_cache_shape(self, batch_size: int, max_length: int) -> Tuple[Tuple[int, ...], Tuple[int, ...]]:
        shape = (batch_size, self.config.n_kv_heads, max_length, self.config.head_dim)
        return shape, shape
