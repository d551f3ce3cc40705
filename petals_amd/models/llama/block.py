"""Native Llama decoder block.

Our own implementation (NOT wrapping transformers — contrast with reference
`models/llama/block.py:226` which wraps HF's LlamaDecoderLayer): RMSNorm ->
GQA attention with RoPE and a preallocated KV cache -> RMSNorm -> SwiGLU MLP.
All hot math goes through `petals_amd.ops` (HIP kernels on MI355X, torch
reference on CPU).

KV cache layout: K and V both [batch, n_kv_heads, max_length, head_dim]
(natural layout; the reference's transposed bloom-layout K is a hivemind wire
convention we do not carry over — conversion happens at the session boundary
if ever needed).

Block forward contract (used by the server backend and tests):
    forward(hidden_states, kv_cache=None, prefix_length=0)
      * kv_cache=None: stateless causal forward over positions [0, seq_len)
      * kv_cache=(k_cache, v_cache): write new K/V at
        [prefix_length : prefix_length + q_len], attend over the full prefix.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import nn

from petals_amd import ops
from petals_amd.models.llama.config import LlamaConfig


class LlamaAttention(nn.Module):
    def __init__(self, config: LlamaConfig):
        super().__init__()
        self.config = config
        self.hidden_size = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.num_kv_heads = config.n_kv_heads
        self.head_dim = config.head_dim
        bias = config.attention_bias
        self.q_proj = nn.Linear(self.hidden_size, self.num_heads * self.head_dim, bias=bias)
        self.k_proj = nn.Linear(self.hidden_size, self.num_kv_heads * self.head_dim, bias=bias)
        self.v_proj = nn.Linear(self.hidden_size, self.num_kv_heads * self.head_dim, bias=bias)
        self.o_proj = nn.Linear(self.num_heads * self.head_dim, self.hidden_size, bias=bias)

        # rope tables are plain (non-module) attributes built lazily on first
        # use: blocks are constructed on the meta device (cheap) and
        # materialized straight on the GPU
        self.rope_cos = None
        self.rope_sin = None

    def _ensure_rope(self, needed_len: int, device, ref: torch.Tensor):
        if self.rope_cos is None or self.rope_cos.shape[0] < needed_len or self.rope_cos.device != torch.device(device):
            prev = 0 if self.rope_cos is None else self.rope_cos.shape[0]
            cos, sin = ops.build_rope_cache(
                self.head_dim,
                max(needed_len, 2 * prev, self.config.max_position_embeddings),
                theta=self.config.rope_theta,
                rope_scaling=self.config.rope_scaling,
            )
            self.rope_cos = cos.to(device)
            self.rope_sin = sin.to(device)

    def forward(
        self,
        hidden_states: torch.Tensor,
        kv_cache: Optional[Tuple[torch.Tensor, torch.Tensor]] = None,
        prefix_length: int = 0,
        adapter=None,  # utils.peft.BlockAdapter
    ) -> torch.Tensor:
        b, q_len, _ = hidden_states.shape

        def proj(lin, key, heads):
            y = lin(hidden_states)
            if adapter is not None:
                d = adapter.delta(key, hidden_states)
                if d is not None:
                    y = y + d
            return y.view(b, q_len, heads, self.head_dim).transpose(1, 2)

        q = proj(self.q_proj, "q", self.num_heads)
        k = proj(self.k_proj, "k", self.num_kv_heads)
        v = proj(self.v_proj, "v", self.num_kv_heads)

        end = prefix_length + q_len
        self._ensure_rope(end, hidden_states.device, hidden_states)
        position_ids = torch.arange(prefix_length, end, device=hidden_states.device)
        q, k = ops.apply_rope(q, k, self.rope_cos, self.rope_sin, position_ids)

        if kv_cache is not None:
            k_cache, v_cache = kv_cache
            k_cache[:b, :, prefix_length:end].copy_(k)
            v_cache[:b, :, prefix_length:end].copy_(v)
            attn = ops.attention_decode(q, k_cache[:b], v_cache[:b], end)
        else:
            assert prefix_length == 0, "stateless forward starts at position 0"
            attn = ops.attention(q, k, v, causal=True)

        attn = attn.transpose(1, 2).reshape(b, q_len, self.num_heads * self.head_dim)
        out = self.o_proj(attn)
        if adapter is not None:
            d = adapter.delta("o", attn)
            if d is not None:
                out = out + d
        return out


class LlamaMLP(nn.Module):
    def __init__(self, config: LlamaConfig):
        super().__init__()
        bias = config.mlp_bias
        self.gate_proj = nn.Linear(config.hidden_size, config.intermediate_size, bias=bias)
        self.up_proj = nn.Linear(config.hidden_size, config.intermediate_size, bias=bias)
        self.down_proj = nn.Linear(config.intermediate_size, config.hidden_size, bias=bias)

    def forward(self, x: torch.Tensor, adapter=None) -> torch.Tensor:
        gate, up = self.gate_proj(x), self.up_proj(x)
        if adapter is not None:
            dg, du = adapter.delta("gate", x), adapter.delta("up", x)
            if dg is not None:
                gate = gate + dg
            if du is not None:
                up = up + du
        act = ops.swiglu(gate, up)
        out = self.down_proj(act)
        if adapter is not None:
            dd = adapter.delta("down", act)
            if dd is not None:
                out = out + dd
        return out


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.variance_epsilon = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.rms_norm(x, self.weight, self.variance_epsilon)


class LlamaBlock(nn.Module):
    """One decoder layer; the unit served by a swarm server."""

    def __init__(self, config: LlamaConfig, layer_idx: int = 0):
        super().__init__()
        self.config = config
        self.layer_idx = layer_idx
        self.self_attn = LlamaAttention(config)
        self.mlp = LlamaMLP(config)
        self.input_layernorm = RMSNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.post_attention_layernorm = RMSNorm(config.hidden_size, eps=config.layer_norm_eps)

    _fast = None  # LlamaFastPath after optimize_for_inference()

    def optimize_for_inference(self, quant: str = "none") -> "LlamaBlock":
        """Repack weights into the MI355X kernel layout (optionally NF4
        quantize-on-load) and enable the fused decode path. Requires the HIP
        extension; frees nn.Linear weights."""
        from petals_amd import ops as _ops
        from petals_amd.ops.fused_decode import LlamaFastPath

        hip = _ops._load_hip_ops()
        if hip is None:
            raise RuntimeError(
                f"cannot optimize block for MI355X: HIP extension missing ({_ops._hip_import_error!r})"
            )
        assert next(self.parameters()).device.type == "cuda", "optimize_for_inference needs a GPU block"
        gq = self.config.num_attention_heads // self.config.n_kv_heads
        if (
            self.config.head_dim not in (64, 128)
            or gq not in (1, 2, 4, 6, 8, 16)
            or self.config.attention_bias
            or getattr(self.config, "mlp_bias", False)
        ):
            import logging

            logging.getLogger(__name__).warning(
                "block geometry (head_dim=%s, gq=%s, bias=%s) outside the fused fast path; "
                "serving via generic HIP ops", self.config.head_dim, gq, self.config.attention_bias,
            )
            return self
        self._fast = LlamaFastPath(self, hip, quant=quant)
        return self

    def forward(
        self,
        hidden_states: torch.Tensor,
        kv_cache: Optional[Tuple[torch.Tensor, torch.Tensor]] = None,
        prefix_length: int = 0,
        ctx=None,  # ops.fused_decode.DecodeContext (device-resident position)
    ) -> torch.Tensor:
        if self._fast is not None:
            from petals_amd.utils.peft import active_block_adapter

            adapter = active_block_adapter(self)
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

        from petals_amd.utils.peft import active_block_adapter

        adapter = active_block_adapter(self)
        residual = hidden_states
        hidden_states = self.input_layernorm(hidden_states)
        hidden_states = self.self_attn(
            hidden_states, kv_cache=kv_cache, prefix_length=prefix_length, adapter=adapter
        )
        hidden_states = residual + hidden_states

        residual = hidden_states
        hidden_states = self.post_attention_layernorm(hidden_states)
        hidden_states = self.mlp(hidden_states, adapter=adapter)
        return residual + hidden_states

    # --- cache geometry used by the server's MemoryCache ---

    def kv_cache_shape(self, batch_size: int, max_length: int) -> Tuple[Tuple[int, ...], Tuple[int, ...]]:
        shape = (batch_size, self.config.n_kv_heads, max_length, self.config.head_dim)
        return shape, shape
