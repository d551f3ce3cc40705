"""Native Falcon decoder block (7B MQA and 40B/180B GQA new-decoder variants).

Weight names match HF Falcon per-layer state dicts. Fused-QKV layouts:
  * new_decoder_architecture: view [b, s, n_kv, n_heads//n_kv + 2, hd] — per
    kv-group: (n_heads//n_kv) query heads, then 1 key and 1 value head
  * multi_query (falcon-7b): view [b, s, n_heads + 2, hd]
(contrast: reference models/falcon/block.py wraps HF FalconDecoderLayer and
CUDA-graphs the QKV split; here the split is plain views and the decode fast
path is the fused HIP kernel.)
"""

from __future__ import annotations

from typing import Tuple

import torch
from torch import nn

from petals_amd import ops
from petals_amd.models.falcon.config import FalconConfig


class FalconAttention(nn.Module):
    def __init__(self, config: FalconConfig):
        super().__init__()
        self.config = config
        self.num_heads = config.num_attention_heads
        self.num_kv = config.n_kv_heads
        self.head_dim = config.head_dim
        h = config.hidden_size
        # shard-safe sizes: a TP shard keeps hidden_size (replicated input)
        # with fewer heads, so derive output widths from heads, not h
        if config.new_decoder_architecture:
            qkv_out = (self.num_heads + 2 * self.num_kv) * self.head_dim
        elif config.multi_query:
            qkv_out = (self.num_heads + 2) * self.head_dim
        else:
            qkv_out = 3 * self.num_heads * self.head_dim
        self.query_key_value = nn.Linear(h, qkv_out, bias=config.bias)
        self.dense = nn.Linear(self.num_heads * self.head_dim, h, bias=config.bias)
        self.rope_cos = None  # lazy (meta-device construction)
        self.rope_sin = None

    def _split_qkv(self, fused: torch.Tensor, b: int, q_len: int):
        if self.config.new_decoder_architecture:
            fused = fused.view(b, q_len, self.num_kv, self.num_heads // self.num_kv + 2, self.head_dim)
            q = fused[..., :-2, :].reshape(b, q_len, self.num_heads, self.head_dim)
            k = fused[..., -2, :]
            v = fused[..., -1, :]
        elif self.config.multi_query:
            fused = fused.view(b, q_len, self.num_heads + 2, self.head_dim)
            q = fused[..., :-2, :]
            k = fused[..., -2:-1, :]
            v = fused[..., -1:, :]
        else:
            fused = fused.view(b, q_len, self.num_heads, 3, self.head_dim)
            q, k, v = fused[..., 0, :], fused[..., 1, :], fused[..., 2, :]
        return (x.transpose(1, 2) for x in (q, k, v))  # [b, heads, len, hd]

    def _ensure_rope(self, needed: int, device):
        if self.rope_cos is None or self.rope_cos.shape[0] < needed or self.rope_cos.device != torch.device(device):
            prev = 0 if self.rope_cos is None else self.rope_cos.shape[0]
            cos, sin = ops.build_rope_cache(
                self.head_dim, max(needed, 2 * prev, self.config.max_position_embeddings), theta=self.config.rope_theta
            )
            self.rope_cos, self.rope_sin = cos.to(device), sin.to(device)

    def forward(self, hidden_states, kv_cache=None, prefix_length: int = 0, adapter=None):
        b, q_len, _ = hidden_states.shape
        fused = self.query_key_value(hidden_states)
        if adapter is not None:
            d = adapter.delta("qkv", hidden_states)
            if d is not None:
                fused = fused + d
        q, k, v = self._split_qkv(fused, b, q_len)
        end = prefix_length + q_len
        self._ensure_rope(end, hidden_states.device)
        pos = torch.arange(prefix_length, end, device=hidden_states.device)
        q, k = ops.apply_rope(q.contiguous(), k.contiguous(), self.rope_cos, self.rope_sin, pos)
        if kv_cache is not None:
            k_cache, v_cache = kv_cache
            k_cache[:b, :, prefix_length:end].copy_(k)
            v_cache[:b, :, prefix_length:end].copy_(v)
            attn = ops.attention_decode(q, k_cache[:b], v_cache[:b], end)
        else:
            assert prefix_length == 0
            attn = ops.attention(q, k, v, causal=True)
        attn = attn.transpose(1, 2).reshape(b, q_len, self.num_heads * self.head_dim)
        out = self.dense(attn)
        if adapter is not None:
            d = adapter.delta("dense", attn)
            if d is not None:
                out = out + d
        return out


class FalconMLP(nn.Module):
    def __init__(self, config: FalconConfig):
        super().__init__()
        self.dense_h_to_4h = nn.Linear(config.hidden_size, config.intermediate_size, bias=config.bias)
        self.dense_4h_to_h = nn.Linear(config.intermediate_size, config.hidden_size, bias=config.bias)

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


class FalconBlock(nn.Module):
    def __init__(self, config: FalconConfig, layer_idx: int = 0):
        super().__init__()
        self.config = config
        self.layer_idx = layer_idx
        self.self_attention = FalconAttention(config)
        self.mlp = FalconMLP(config)
        eps = config.layer_norm_eps
        if config.new_decoder_architecture:
            self.ln_attn = nn.LayerNorm(config.hidden_size, eps=eps)
            self.ln_mlp = nn.LayerNorm(config.hidden_size, eps=eps)
        else:
            self.input_layernorm = nn.LayerNorm(config.hidden_size, eps=eps)
            if not config.parallel_attn:
                self.post_attention_layernorm = nn.LayerNorm(config.hidden_size, eps=eps)

    _fast = None  # FalconFastPath after optimize_for_inference()
    tp_world = 1  # set by parallel/tp.py TPFalconBlock shards
    tp_group = None

    # (head_dim, gq) combos instantiated in ops/csrc/attention.hip ATTN_CASE /
    # ATTN_CASE_BIG — gq > 16 runs the multi-group MFMA decode (ceil(gq/16)
    # A-fragment groups per block, K/V re-read per extra group)
    _FUSED_DECODE_GEOMS = {
        (128, 1), (128, 2), (128, 4), (128, 6), (128, 8), (128, 16), (128, 32),
        (64, 1), (64, 2), (64, 4), (64, 8), (64, 16), (64, 29), (64, 32), (64, 71),
    }

    def optimize_for_inference(self, quant: str = "none") -> "FalconBlock":
        """Repack weights into the MI355X kernel layout (parallel attn+MLP,
        rope GQA fused decode chain); frees nn.Linear weights. Covers the
        new-decoder architecture (40B/180B GQA) and the 7B-style old decoder
        (MQA + parallel_attn, single layernorm); other variants (e.g. biased
        falcon-rw) serve via generic ops."""
        from petals_amd import ops as _ops
        from petals_amd.ops.fused_decode import FalconFastPath

        hip = _ops._load_hip_ops()
        if hip is None:
            raise RuntimeError(
                f"cannot optimize block for MI355X: HIP extension missing ({_ops._hip_import_error!r})"
            )
        assert next(self.parameters()).device.type == "cuda", "optimize_for_inference needs a GPU block"
        cfg = self.config
        gq = cfg.num_attention_heads // cfg.n_kv_heads
        if (
            (cfg.head_dim, gq) not in self._FUSED_DECODE_GEOMS
            or (not cfg.new_decoder_architecture and not cfg.parallel_attn)
            or cfg.bias
        ):
            import logging

            logging.getLogger(__name__).warning(
                "Falcon geometry (new_decoder=%s, parallel_attn=%s, head_dim=%s, gq=%s, bias=%s) "
                "outside the fused fast path; serving via generic HIP ops",
                cfg.new_decoder_architecture, cfg.parallel_attn, cfg.head_dim, gq, cfg.bias,
            )
            return self
        self._fast = FalconFastPath(self, hip, quant=quant, tp_world=self.tp_world, tp_group=self.tp_group)
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

        residual = hidden_states
        if self.config.new_decoder_architecture:
            attn_in = self.ln_attn(hidden_states)
            mlp_in = self.ln_mlp(hidden_states)
            attn = self.self_attention(attn_in, kv_cache=kv_cache, prefix_length=prefix_length, adapter=adapter)
            return residual + attn + self.mlp(mlp_in, adapter=adapter)
        attn_in = self.input_layernorm(hidden_states)
        attn = self.self_attention(attn_in, kv_cache=kv_cache, prefix_length=prefix_length, adapter=adapter)
        if self.config.parallel_attn:
            return residual + attn + self.mlp(attn_in, adapter=adapter)
        hidden_states = residual + attn
        mlp_in = self.post_attention_layernorm(hidden_states)
        return hidden_states + self.mlp(mlp_in, adapter=adapter)

    def kv_cache_shape(self, batch_size: int, max_length: int) -> Tuple[Tuple[int, ...], Tuple[int, ...]]:
        shape = (batch_size, self.config.n_kv_heads, max_length, self.config.head_dim)
        return shape, shape
