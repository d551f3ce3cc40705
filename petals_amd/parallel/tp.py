"""Tensor parallelism: rank-per-GPU block shards with RCCL all-reduce.

The MI355X-native replacement for the reference's `tensor_parallel` package
(single CUDA process driving N devices, utils/convert_block.py:118-135): here
every GPU is its own process (one rank per GPU, torch.distributed over
RCCL/xGMI), a block is sharded column-parallel for QKV/gate/up (split by
heads / intermediate columns) and row-parallel for O/down, with one
all-reduce(SUM) after attention and one after the MLP — reduce traffic is
2 x hidden per token per block, bucketed by RCCL over the 7-link xGMI mesh.

KV caches are per-rank shards ([batch, kv_heads/world, len, head_dim]) — the
reference's PerDeviceTensors equivalent. Requires kv_heads % world == 0.

Works on CPU with the gloo backend for tests (world 2), exactly like the
reference's `--tensor_parallel_devices cpu cpu` CI servers.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist
from torch import nn

from petals_amd import ops
from petals_amd.models.llama.block import RMSNorm
from petals_amd.models.llama.config import LlamaConfig


def _all_reduce(t: torch.Tensor, group=None) -> torch.Tensor:
    if dist.is_initialized() and dist.get_world_size(group) > 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=group)
    return t


class TPLlamaAttention(nn.Module):
    def __init__(self, config: LlamaConfig, rank: int, world: int, group=None):
        super().__init__()
        assert config.num_attention_heads % world == 0, "q heads must divide world"
        assert config.n_kv_heads % world == 0, "kv heads must divide world (raise TP degree granularity)"
        self.config = config
        self.rank, self.world, self.group = rank, world, group
        self.num_heads = config.num_attention_heads // world
        self.num_kv_heads = config.n_kv_heads // world
        self.head_dim = config.head_dim
        h = config.hidden_size
        bias = config.attention_bias
        self.q_proj = nn.Linear(h, self.num_heads * self.head_dim, bias=bias)
        self.k_proj = nn.Linear(h, self.num_kv_heads * self.head_dim, bias=bias)
        self.v_proj = nn.Linear(h, self.num_kv_heads * self.head_dim, bias=bias)
        self.o_proj = nn.Linear(self.num_heads * self.head_dim, h, bias=False)  # row-parallel: bias once
        self.rope_cos = None
        self.rope_sin = None

    def _ensure_rope(self, needed: int, device):
        if self.rope_cos is None or self.rope_cos.shape[0] < needed or self.rope_cos.device != torch.device(device):
            cos, sin = ops.build_rope_cache(
                self.head_dim, max(needed, self.config.max_position_embeddings),
                theta=self.config.rope_theta, rope_scaling=self.config.rope_scaling,
            )
            self.rope_cos, self.rope_sin = cos.to(device), sin.to(device)

    def forward(self, x, kv_cache=None, prefix_length: int = 0):
        b, q_len, _ = x.shape
        q = self.q_proj(x).view(b, q_len, self.num_heads, self.head_dim).transpose(1, 2)
        k = self.k_proj(x).view(b, q_len, self.num_kv_heads, self.head_dim).transpose(1, 2)
        v = self.v_proj(x).view(b, q_len, self.num_kv_heads, self.head_dim).transpose(1, 2)
        end = prefix_length + q_len
        self._ensure_rope(end, x.device)
        pos = torch.arange(prefix_length, end, device=x.device)
        q, k = ops.apply_rope(q, k, self.rope_cos, self.rope_sin, pos)
        if kv_cache is not None:
            k_cache, v_cache = kv_cache  # per-rank shard caches
            k_cache[:b, :, prefix_length:end].copy_(k)
            v_cache[:b, :, prefix_length:end].copy_(v)
            attn = ops.attention_decode(q, k_cache[:b], v_cache[:b], end)
        else:
            assert prefix_length == 0
            attn = ops.attention(q, k, v, causal=True)
        attn = attn.transpose(1, 2).reshape(b, q_len, self.num_heads * self.head_dim)
        out = self.o_proj(attn)  # partial sum
        return _all_reduce(out, self.group)


class TPLlamaMLP(nn.Module):
    def __init__(self, config: LlamaConfig, rank: int, world: int, group=None):
        super().__init__()
        assert config.intermediate_size % world == 0
        inter = config.intermediate_size // world
        self.group = group
        self.gate_proj = nn.Linear(config.hidden_size, inter, bias=False)
        self.up_proj = nn.Linear(config.hidden_size, inter, bias=False)
        self.down_proj = nn.Linear(inter, config.hidden_size, bias=False)

    def forward(self, x):
        out = self.down_proj(ops.swiglu(self.gate_proj(x), self.up_proj(x)))
        return _all_reduce(out, self.group)


class TPLlamaBlock(nn.Module):
    """One rank's shard of a Llama decoder block."""

    def __init__(self, config: LlamaConfig, layer_idx: int = 0, rank: Optional[int] = None,
                 world: Optional[int] = None, group=None):
        super().__init__()
        if rank is None:
            rank = dist.get_rank(group) if dist.is_initialized() else 0
        if world is None:
            world = dist.get_world_size(group) if dist.is_initialized() else 1
        self.config = config
        self.layer_idx = layer_idx
        self.rank, self.world = rank, world
        self.self_attn = TPLlamaAttention(config, rank, world, group)
        self.mlp = TPLlamaMLP(config, rank, world, group)
        self.input_layernorm = RMSNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.post_attention_layernorm = RMSNorm(config.hidden_size, eps=config.layer_norm_eps)

    def load_from_full_state_dict(self, sd: dict) -> None:
        """Slice a FULL block state dict into this rank's shard (column split
        by heads for q/k/v and by columns for gate/up; row split for o/down)."""
        r, w = self.rank, self.world
        hd = self.config.head_dim

        def rows(t, n_shard_rows):
            return t[r * n_shard_rows : (r + 1) * n_shard_rows]

        def cols(t, n_shard_cols):
            return t[:, r * n_shard_cols : (r + 1) * n_shard_cols]

        a = self.self_attn
        a.q_proj.weight.data.copy_(rows(sd["self_attn.q_proj.weight"], a.num_heads * hd))
        a.k_proj.weight.data.copy_(rows(sd["self_attn.k_proj.weight"], a.num_kv_heads * hd))
        a.v_proj.weight.data.copy_(rows(sd["self_attn.v_proj.weight"], a.num_kv_heads * hd))
        a.o_proj.weight.data.copy_(cols(sd["self_attn.o_proj.weight"], a.num_heads * hd))
        m = self.mlp
        inter = m.gate_proj.out_features
        m.gate_proj.weight.data.copy_(rows(sd["mlp.gate_proj.weight"], inter))
        m.up_proj.weight.data.copy_(rows(sd["mlp.up_proj.weight"], inter))
        m.down_proj.weight.data.copy_(cols(sd["mlp.down_proj.weight"], inter))
        self.input_layernorm.weight.data.copy_(sd["input_layernorm.weight"])
        self.post_attention_layernorm.weight.data.copy_(sd["post_attention_layernorm.weight"])

    def forward(self, hidden_states, kv_cache=None, prefix_length: int = 0, ctx=None):
        residual = hidden_states
        hidden_states = self.input_layernorm(hidden_states)
        hidden_states = self.self_attn(hidden_states, kv_cache=kv_cache, prefix_length=prefix_length)
        hidden_states = residual + hidden_states
        residual = hidden_states
        hidden_states = self.post_attention_layernorm(hidden_states)
        hidden_states = self.mlp(hidden_states)
        return residual + hidden_states

    def kv_cache_shape(self, batch_size: int, max_length: int) -> Tuple[Tuple[int, ...], Tuple[int, ...]]:
        shape = (batch_size, self.self_attn.num_kv_heads, max_length, self.config.head_dim)
        return shape, shape
