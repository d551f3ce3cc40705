"""Native BLOOM decoder block: LayerNorm -> fused-QKV attention with ALiBi ->
LayerNorm -> GELU MLP. Weight names match HF BLOOM per-layer state dicts
(contrast: reference models/bloom/block.py wraps HF's BloomBlock).

HF fused-QKV layout: query_key_value output reshapes to
[batch, seq, n_heads, 3, head_dim] (per-head interleave of q,k,v).
"""

from __future__ import annotations

from typing import Tuple

import torch
from torch import nn

from petals_amd import ops
from petals_amd.models.bloom.config import BloomConfig


class BloomAttention(nn.Module):
    def __init__(self, config: BloomConfig):
        super().__init__()
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.hidden_size = config.hidden_size
        # shard-safe sizes: for a full model num_heads*head_dim == hidden_size;
        # a TP shard keeps hidden_size (replicated input) with fewer heads
        qkv_out = 3 * self.num_heads * self.head_dim
        self.query_key_value = nn.Linear(self.hidden_size, qkv_out, bias=True)
        self.dense = nn.Linear(self.num_heads * self.head_dim, self.hidden_size, bias=True)
        # ALiBi slopes depend on the GLOBAL head index: TP shard configs carry
        # their slice window (alibi_start_head / alibi_total_heads)
        self._alibi_total = getattr(config, "alibi_total_heads", None) or self.num_heads
        self._alibi_start = getattr(config, "alibi_start_head", 0)
        self._alibi_slopes = None  # lazy: blocks may be built on the meta device

    def forward(self, hidden_states, kv_cache=None, prefix_length: int = 0, adapter=None):
        b, q_len, _ = hidden_states.shape
        fused = self.query_key_value(hidden_states)
        if adapter is not None:
            d = adapter.delta("qkv", hidden_states)
            if d is not None:
                fused = fused + d
        fused = fused.view(b, q_len, self.num_heads, 3, self.head_dim)
        q = fused[..., 0, :].transpose(1, 2)  # [b, heads, q_len, hd]
        k = fused[..., 1, :].transpose(1, 2)
        v = fused[..., 2, :].transpose(1, 2)

        end = prefix_length + q_len
        if self._alibi_slopes is None or self._alibi_slopes.device != hidden_states.device:
            full = ops.build_alibi_slopes(self._alibi_total)
            self._alibi_slopes = full[self._alibi_start : self._alibi_start + self.num_heads].to(
                hidden_states.device
            )
        slopes = self._alibi_slopes
        if kv_cache is not None:
            k_cache, v_cache = kv_cache
            k_cache[:b, :, prefix_length:end].copy_(k)
            v_cache[:b, :, prefix_length:end].copy_(v)
            attn = ops.attention_decode(q, k_cache[:b], v_cache[:b], end, alibi_slopes=slopes)
        else:
            assert prefix_length == 0
            attn = ops.attention(q, k, v, causal=True, alibi_slopes=slopes)
        attn = attn.transpose(1, 2).reshape(b, q_len, self.num_heads * self.head_dim)
        out = self.dense(attn)
        if adapter is not None:
            d = adapter.delta("dense", attn)
            if d is not None:
                out = out + d
        return out


class BloomMLP(nn.Module):
    def __init__(self, config: BloomConfig):
        super().__init__()
        self.dense_h_to_4h = nn.Linear(config.hidden_size, config.intermediate_size, bias=True)
        self.dense_4h_to_h = nn.Linear(config.intermediate_size, config.hidden_size, bias=True)

    def forward(self, x, adapter=None):
        inter = self.dense_h_to_4h(x)
        if adapter is not None:
            d = adapter.delta("h4h", x)
            if d is not None:
                inter = inter + d
        act = ops.gelu(inter)
        out = self.dense_4h_to_h(act)
        if adapter is not None:
            d = adapter.delta("4hh", act)
            if d is not None:
                out = out + d
        return out


class BloomBlock(nn.Module):
    def __init__(self, config: BloomConfig, layer_idx: int = 0):
        super().__init__()
        self.config = config
        self.layer_idx = layer_idx
        self.input_layernorm = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.self_attention = BloomAttention(config)
        self.post_attention_layernorm = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.mlp = BloomMLP(config)
        self.apply_residual_post_ln = config.apply_residual_connection_post_layernorm

    _fast = None  # BloomFastPath after optimize_for_inference()
    tp_world = 1  # set by parallel/tp.py TPBloomBlock shards
    tp_group = None

    def optimize_for_inference(self, quant: str = "none") -> "BloomBlock":
        """Repack weights into the MI355X kernel layout (LayerNorm + ALiBi +
        GELU fused decode chain); frees nn.Linear weights."""
        from petals_amd import ops as _ops
        from petals_amd.ops.fused_decode import BloomFastPath

        hip = _ops._load_hip_ops()
        if hip is None:
            raise RuntimeError(
                f"cannot optimize block for MI355X: HIP extension missing ({_ops._hip_import_error!r})"
            )
        assert next(self.parameters()).device.type == "cuda", "optimize_for_inference needs a GPU block"
        if self.config.head_dim not in (64, 128):
            import logging

            logging.getLogger(__name__).warning(
                "BLOOM head_dim=%s outside the fused fast path; serving via generic HIP ops",
                self.config.head_dim,
            )
            return self
        self._fast = BloomFastPath(self, hip, quant=quant, tp_world=self.tp_world, tp_group=self.tp_group)
        return self

    def forward(self, hidden_states, kv_cache=None, prefix_length: int = 0, ctx=None):
        from petals_amd.utils.peft import active_block_adapter

        adapter = active_block_adapter(self)
        if self._fast is not None:
            if torch.is_grad_enabled() and hidden_states.requires_grad:
                assert kv_cache is None, "training forward does not use the KV cache"
                return self._fast.forward_autograd(hidden_states, prefix_length, adapter=adapter)
            if kv_cache is not None and hidden_states.shape[1] == 1 and hidden_states.shape[0] <= 8:
                from petals_amd.ops.fused_decode import decode_step_auto

                return decode_step_auto(
                    self._fast, hidden_states, kv_cache[0], kv_cache[1], prefix_length,
                    ctx=ctx, adapter=adapter,
                )
            return self._fast.forward(hidden_states, kv_cache, prefix_length, adapter=adapter)

        ln_out = self.input_layernorm(hidden_states)
        residual = ln_out if self.apply_residual_post_ln else hidden_states
        attn = self.self_attention(ln_out, kv_cache=kv_cache, prefix_length=prefix_length, adapter=adapter)
        hidden_states = residual + attn

        ln_out = self.post_attention_layernorm(hidden_states)
        residual = ln_out if self.apply_residual_post_ln else hidden_states
        return residual + self.mlp(ln_out, adapter=adapter)

    def kv_cache_shape(self, batch_size: int, max_length: int) -> Tuple[Tuple[int, ...], Tuple[int, ...]]:
        shape = (batch_size, self.config.num_attention_heads, max_length, self.config.head_dim)
        return shape, shape
