"""Tensor parallelism: rank-per-GPU block shards with RCCL all-reduce.

The MI355X-native replacement for the reference's `tensor_parallel` package
(single CUDA process driving N devices, utils/convert_block.py:118-135): here
every GPU is its own process (one rank per GPU, torch.distributed over
RCCL/xGMI), a block is sharded column-parallel for QKV/gate/up (split by
heads / intermediate columns) and row-parallel for O/down, with one
all-reduce(SUM) after attention and one after the MLP — reduce traffic is
2 x hidden per token per block over the 7-link xGMI mesh.

The fused MI355X decode path applies unchanged to a shard: a TP block exposes
its SHARD geometry as `block.config`, so `optimize_for_inference()` builds the
same LlamaFastPath (NF4/int8 gemv, MFMA flash decode, fused epilogues) over
the shard weights, with the residual-fused epilogues split at the two reduce
points (partial -> all_reduce -> +residual).

Training: column-parallel layers consume the replicated input (grad wrt input
needs an all-reduce in backward — `copy_to_tp`), row-parallel partial sums are
reduced in forward with identity backward (`reduce_from_tp`).

KV caches are per-rank shards ([batch, kv_heads/world, len, head_dim]) — the
reference's PerDeviceTensors equivalent. Requires kv_heads % world == 0.

Works on CPU with the gloo backend for tests (world 2), exactly like the
reference's `--tensor_parallel_devices cpu cpu` CI servers.
"""

from __future__ import annotations

import dataclasses
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


class _CopyToTP(torch.autograd.Function):
    """Identity forward; all-reduce(SUM) backward — wraps the replicated input
    of column-parallel layers so input grads combine every rank's shard."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        return _all_reduce(grad.contiguous(), ctx.group), None


class _ReduceFromTP(torch.autograd.Function):
    """All-reduce(SUM) forward (combine row-parallel partials); identity
    backward."""

    @staticmethod
    def forward(ctx, x, group):
        return _all_reduce(x.contiguous(), group)

    @staticmethod
    def backward(ctx, grad):
        return grad, None


def copy_to_tp(x, group=None):
    return _CopyToTP.apply(x, group) if torch.is_grad_enabled() and x.requires_grad else x


def reduce_from_tp(x, group=None):
    if torch.is_grad_enabled() and x.requires_grad:
        return _ReduceFromTP.apply(x, group)
    return _all_reduce(x, group)


def shard_llama_config(config: LlamaConfig, world: int) -> LlamaConfig:
    """This rank's shard geometry as a plain config (the fused fast path reads
    its shapes from here)."""
    assert config.num_attention_heads % world == 0, "q heads must divide tp world"
    assert config.n_kv_heads % world == 0, "kv heads must divide tp world"
    assert config.intermediate_size % world == 0
    return dataclasses.replace(
        config,
        num_attention_heads=config.num_attention_heads // world,
        num_key_value_heads=config.n_kv_heads // world,
        intermediate_size=config.intermediate_size // world,
        head_dim_override=config.head_dim,
    )


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
        self.scale = None  # default 1/sqrt(head_dim) inside ops

    def _ensure_rope(self, needed: int, device):
        if self.rope_cos is None or self.rope_cos.shape[0] < needed or self.rope_cos.device != torch.device(device):
            cos, sin = ops.build_rope_cache(
                self.head_dim, max(needed, self.config.max_position_embeddings),
                theta=self.config.rope_theta, rope_scaling=self.config.rope_scaling,
            )
            self.rope_cos, self.rope_sin = cos.to(device), sin.to(device)

    def forward(self, x, kv_cache=None, prefix_length: int = 0):
        # copy boundary: grads of the replicated input through THIS rank's
        # column shard are partial -> all-reduced in backward. (Residual and
        # layernorm paths are replicated and must NOT be reduced.)
        x = copy_to_tp(x, self.group)
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
        return reduce_from_tp(out, self.group)


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
        x = copy_to_tp(x, self.group)
        out = self.down_proj(ops.swiglu(self.gate_proj(x), self.up_proj(x)))
        return reduce_from_tp(out, self.group)


class TPLlamaBlock(nn.Module):
    """One rank's shard of a Llama decoder block.

    `self.config` reports the SHARD geometry, so `optimize_for_inference()`
    can wrap the shard in the SAME fused MI355X path (LlamaFastPath: NF4/int8
    gemv, MFMA flash decode) as an unsharded block — with the two
    residual-fused epilogues split at the all-reduce points."""

    def __init__(self, config: LlamaConfig, layer_idx: int = 0, rank: Optional[int] = None,
                 world: Optional[int] = None, group=None):
        super().__init__()
        if rank is None:
            rank = dist.get_rank(group) if dist.is_initialized() else 0
        if world is None:
            world = dist.get_world_size(group) if dist.is_initialized() else 1
        self.full_config = config
        self.config = shard_llama_config(config, world)  # shard geometry
        self.layer_idx = layer_idx
        self.rank, self.world = rank, world
        self.tp_group = group
        self.self_attn = TPLlamaAttention(config, rank, world, group)
        self.mlp = TPLlamaMLP(config, rank, world, group)
        self.input_layernorm = RMSNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.post_attention_layernorm = RMSNorm(config.hidden_size, eps=config.layer_norm_eps)

    _fast = None  # LlamaFastPath over the shard, after optimize_for_inference()

    def optimize_for_inference(self, quant: str = "none") -> "TPLlamaBlock":
        from petals_amd import ops as _ops
        from petals_amd.ops.fused_decode import LlamaFastPath

        hip = _ops._load_hip_ops()
        if hip is None:
            raise RuntimeError(
                f"cannot optimize block for MI355X: HIP extension missing ({_ops._hip_import_error!r})"
            )
        assert next(self.parameters()).device.type == "cuda", "optimize_for_inference needs a GPU block"
        self._fast = LlamaFastPath(self, hip, quant=quant, tp_world=self.world, tp_group=self.tp_group)
        return self

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
        if self._fast is not None:
            if torch.is_grad_enabled() and hidden_states.requires_grad:
                assert kv_cache is None, "training forward does not use the KV cache"
                return self._fast.forward_autograd(hidden_states, prefix_length)
            if kv_cache is not None and hidden_states.shape[1] == 1 and hidden_states.shape[0] <= 8:
                from petals_amd.ops.fused_decode import decode_step_auto

                return decode_step_auto(
                    self._fast, hidden_states, kv_cache[0], kv_cache[1], prefix_length, ctx=ctx
                )
            return self._fast.forward(hidden_states, kv_cache, prefix_length)
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


# ---------------------------------------------------------------------------
# Falcon / BLOOM tensor parallelism. Unlike llama, these families shard
# UNEVENLY when the head count does not divide the TP degree (falcon-7b has a
# PRIME 71 query heads): rank r takes a contiguous near-even slice of heads
# (or of kv GROUPS for the falcon new-decoder architecture, whose fused-QKV
# weight is group-major), so per-rank shard configs differ in head count but
# all produce [*, hidden] row-parallel partials for the same all-reduce.
# Falcon MQA replicates the single k/v head on every rank (its weight rows are
# loaded whole); parallel-attention falcons need only ONE all-reduce per block
# (attn partial + mlp partial summed locally first).
# ---------------------------------------------------------------------------


def _split_count(total: int, world: int, rank: int) -> Tuple[int, int]:
    """Contiguous near-even partition: (start, count) for this rank."""
    base, rem = divmod(total, world)
    start = rank * base + min(rank, rem)
    return start, base + (1 if rank < rem else 0)


def shard_falcon_config(config, world: int, rank: int):
    from petals_amd.models.falcon.config import FalconConfig  # noqa: F401

    i_start, i_n = _split_count(config.intermediate_size, world, rank)
    if config.new_decoder_architecture:
        gq = config.num_attention_heads // config.n_kv_heads
        g_start, g_n = _split_count(config.n_kv_heads, world, rank)
        return dataclasses.replace(
            config,
            num_attention_heads=g_n * gq,
            num_kv_heads=g_n,
            intermediate_size=i_n,
            head_dim_override=config.head_dim,
        )
    q_start, q_n = _split_count(config.num_attention_heads, world, rank)
    return dataclasses.replace(
        config,
        num_attention_heads=q_n,
        intermediate_size=i_n,
        head_dim_override=config.head_dim,
    )  # __post_init__ re-derives num_key_value_heads (1 for MQA, =heads classic)


def shard_bloom_config(config, world: int, rank: int):
    h_start, h_n = _split_count(config.num_attention_heads, world, rank)
    i_start, i_n = _split_count(config.intermediate_size, world, rank)
    return dataclasses.replace(
        config,
        num_attention_heads=h_n,
        num_key_value_heads=h_n,
        intermediate_size=i_n,
        head_dim_override=config.head_dim,
        alibi_start_head=h_start,
        alibi_total_heads=config.num_attention_heads,
    )


def _resolve_rank_world(rank, world, group):
    if rank is None:
        rank = dist.get_rank(group) if dist.is_initialized() else 0
    if world is None:
        world = dist.get_world_size(group) if dist.is_initialized() else 1
    return rank, world


from petals_amd.models.bloom.block import BloomBlock as _BloomBlockBase  # noqa: E402
from petals_amd.models.falcon.block import FalconBlock as _FalconBlockBase  # noqa: E402


class TPFalconBlock(_FalconBlockBase):
    """One rank's shard of a Falcon decoder block (all three architectures:
    new-decoder GQA 40B/180B, MQA 7B, classic MHA rw). `self.config` reports
    the SHARD geometry so `optimize_for_inference()` builds the same fused
    FalconFastPath over the shard (tp hooks split the residual epilogue at the
    single per-block all-reduce)."""

    def __init__(self, config, layer_idx: int = 0, rank: Optional[int] = None,
                 world: Optional[int] = None, group=None):
        rank, world = _resolve_rank_world(rank, world, group)
        shard = shard_falcon_config(config, world, rank)
        super().__init__(shard, layer_idx)
        self.full_config = config
        self.rank, self.world = rank, world
        self.tp_world, self.tp_group = world, group

    def load_from_full_state_dict(self, sd: dict) -> None:
        cfg, full = self.config, self.full_config
        r, w, hd = self.rank, self.world, cfg.head_dim
        qkv_w = sd["self_attention.query_key_value.weight"]
        qkv_b = sd.get("self_attention.query_key_value.bias")
        if full.new_decoder_architecture:
            gq = full.num_attention_heads // full.n_kv_heads
            g_start, g_n = _split_count(full.n_kv_heads, w, r)
            rows = slice(g_start * (gq + 2) * hd, (g_start + g_n) * (gq + 2) * hd)
            cols = slice(g_start * gq * hd, (g_start + g_n) * gq * hd)
            qkv_rows_w = qkv_w[rows]
            qkv_rows_b = qkv_b[rows] if qkv_b is not None else None
        elif full.multi_query:
            q_start, q_n = _split_count(full.num_attention_heads, w, r)
            qrows = slice(q_start * hd, (q_start + q_n) * hd)
            kvrows = slice(full.num_attention_heads * hd, (full.num_attention_heads + 2) * hd)
            qkv_rows_w = torch.cat([qkv_w[qrows], qkv_w[kvrows]], dim=0)  # q shard + replicated k,v
            qkv_rows_b = torch.cat([qkv_b[qrows], qkv_b[kvrows]], dim=0) if qkv_b is not None else None
            cols = qrows
        else:  # classic MHA: per-head [q,k,v] interleave, 3*hd rows per head
            h_start, h_n = _split_count(full.num_attention_heads, w, r)
            rows = slice(h_start * 3 * hd, (h_start + h_n) * 3 * hd)
            cols = slice(h_start * hd, (h_start + h_n) * hd)
            qkv_rows_w = qkv_w[rows]
            qkv_rows_b = qkv_b[rows] if qkv_b is not None else None
        a = self.self_attention
        a.query_key_value.weight.data.copy_(qkv_rows_w)
        a.dense.weight.data.copy_(sd["self_attention.dense.weight"][:, cols])
        if full.bias:
            a.query_key_value.bias.data.copy_(qkv_rows_b)
            # row-parallel biases apply ONCE: rank 0 carries them, others zero
            a.dense.bias.data.copy_(sd["self_attention.dense.bias"]) if r == 0 else a.dense.bias.data.zero_()
        i_start, i_n = _split_count(full.intermediate_size, w, r)
        irows = slice(i_start, i_start + i_n)
        m = self.mlp
        m.dense_h_to_4h.weight.data.copy_(sd["mlp.dense_h_to_4h.weight"][irows])
        m.dense_4h_to_h.weight.data.copy_(sd["mlp.dense_4h_to_h.weight"][:, irows])
        if full.bias:
            m.dense_h_to_4h.bias.data.copy_(sd["mlp.dense_h_to_4h.bias"][irows])
            m.dense_4h_to_h.bias.data.copy_(sd["mlp.dense_4h_to_h.bias"]) if r == 0 else m.dense_4h_to_h.bias.data.zero_()
        for name in ("ln_attn", "ln_mlp", "input_layernorm", "post_attention_layernorm"):
            mod = getattr(self, name, None)
            if mod is not None:
                mod.weight.data.copy_(sd[f"{name}.weight"])
                mod.bias.data.copy_(sd[f"{name}.bias"])

    def forward(self, hidden_states, kv_cache=None, prefix_length: int = 0, ctx=None):
        if self._fast is not None:  # fused path reduces internally (tp hooks)
            return super().forward(hidden_states, kv_cache=kv_cache, prefix_length=prefix_length, ctx=ctx)
        g = self.tp_group
        residual = hidden_states
        if self.config.new_decoder_architecture:
            attn_in = copy_to_tp(self.ln_attn(hidden_states), g)
            mlp_in = copy_to_tp(self.ln_mlp(hidden_states), g)
            part = self.self_attention(attn_in, kv_cache=kv_cache, prefix_length=prefix_length) + self.mlp(mlp_in)
            return residual + reduce_from_tp(part, g)  # ONE reduce (parallel residual)
        attn_in = copy_to_tp(self.input_layernorm(hidden_states), g)
        attn = self.self_attention(attn_in, kv_cache=kv_cache, prefix_length=prefix_length)
        if self.config.parallel_attn:
            return residual + reduce_from_tp(attn + self.mlp(attn_in), g)
        hidden_states = residual + reduce_from_tp(attn, g)
        mlp_in = copy_to_tp(self.post_attention_layernorm(hidden_states), g)
        return hidden_states + reduce_from_tp(self.mlp(mlp_in), g)


class TPBloomBlock(_BloomBlockBase):
    """One rank's shard of a BLOOM decoder block: heads sharded (ALiBi slopes
    sliced by GLOBAL head index via the shard config's alibi window)."""

    def __init__(self, config, layer_idx: int = 0, rank: Optional[int] = None,
                 world: Optional[int] = None, group=None):
        rank, world = _resolve_rank_world(rank, world, group)
        shard = shard_bloom_config(config, world, rank)
        super().__init__(shard, layer_idx)
        self.full_config = config
        self.rank, self.world = rank, world
        self.tp_world, self.tp_group = world, group

    def load_from_full_state_dict(self, sd: dict) -> None:
        full = self.full_config
        r, w, hd = self.rank, self.world, self.config.head_dim
        h_start, h_n = _split_count(full.num_attention_heads, w, r)
        rows = slice(h_start * 3 * hd, (h_start + h_n) * 3 * hd)  # per-head [q,k,v] interleave
        cols = slice(h_start * hd, (h_start + h_n) * hd)
        a = self.self_attention
        a.query_key_value.weight.data.copy_(sd["self_attention.query_key_value.weight"][rows])
        a.query_key_value.bias.data.copy_(sd["self_attention.query_key_value.bias"][rows])
        a.dense.weight.data.copy_(sd["self_attention.dense.weight"][:, cols])
        # row-parallel biases apply ONCE: rank 0 carries them, others zero
        a.dense.bias.data.copy_(sd["self_attention.dense.bias"]) if r == 0 else a.dense.bias.data.zero_()
        i_start, i_n = _split_count(full.intermediate_size, w, r)
        irows = slice(i_start, i_start + i_n)
        m = self.mlp
        m.dense_h_to_4h.weight.data.copy_(sd["mlp.dense_h_to_4h.weight"][irows])
        m.dense_h_to_4h.bias.data.copy_(sd["mlp.dense_h_to_4h.bias"][irows])
        m.dense_4h_to_h.weight.data.copy_(sd["mlp.dense_4h_to_h.weight"][:, irows])
        m.dense_4h_to_h.bias.data.copy_(sd["mlp.dense_4h_to_h.bias"]) if r == 0 else m.dense_4h_to_h.bias.data.zero_()
        for name in ("input_layernorm", "post_attention_layernorm"):
            mod = getattr(self, name)
            mod.weight.data.copy_(sd[f"{name}.weight"])
            mod.bias.data.copy_(sd[f"{name}.bias"])

    def forward(self, hidden_states, kv_cache=None, prefix_length: int = 0, ctx=None):
        if self._fast is not None:  # fused path reduces internally (tp hooks)
            return super().forward(hidden_states, kv_cache=kv_cache, prefix_length=prefix_length, ctx=ctx)
        g = self.tp_group
        ln_out = self.input_layernorm(hidden_states)
        res = ln_out if self.apply_residual_post_ln else hidden_states
        attn = self.self_attention(copy_to_tp(ln_out, g), kv_cache=kv_cache, prefix_length=prefix_length)
        h2 = res + reduce_from_tp(attn, g)
        ln2 = self.post_attention_layernorm(h2)
        res2 = ln2 if self.apply_residual_post_ln else h2
        return res2 + reduce_from_tp(self.mlp(copy_to_tp(ln2, g)), g)


def build_tp_block(config, layer_idx: int, *, rank: int, world: int, group=None) -> nn.Module:
    """TP shard block factory (llama GQA+SwiGLU, falcon all three
    architectures, bloom MHA+ALiBi). Other families (e.g. Mixtral, whose
    experts parallelize better by expert than by column) serve single-rank."""
    if config.model_type in ("llama",):
        return TPLlamaBlock(config, layer_idx, rank=rank, world=world, group=group)
    if config.model_type == "falcon":
        return TPFalconBlock(config, layer_idx, rank=rank, world=world, group=group)
    if config.model_type == "bloom":
        return TPBloomBlock(config, layer_idx, rank=rank, world=world, group=group)
    raise NotImplementedError(f"tensor parallelism is not implemented for model_type={config.model_type!r}")


# ---------------------------------------------------------------------------
# TP serving coordination: rank 0 runs the FULL swarm Server (DHT, handler,
# runtime); ranks 1..N-1 run TPShadowWorker loops that execute their block
# shards in lockstep. Every coordinator op is two broadcasts on the tp group
# (a small int64 control tensor, then payload tensors); rank0 serializes ops
# under a lock so the collective order is identical on every rank.
# ---------------------------------------------------------------------------

OP_SHUTDOWN, OP_STEP, OP_OPEN, OP_CLOSE, OP_FWD, OP_BWD, OP_SPAN = 0, 1, 2, 3, 4, 5, 6
_CTRL_LEN = 8


class TPCoordinator:
    """Rank-0 side: broadcast work to the shadow ranks, then run the same
    compute locally (the collectives inside the blocks pair up)."""

    def __init__(self, group, device, hidden_size: int, dtype):
        import threading

        self.group = group
        self.device = torch.device(device)
        self.hidden_size = hidden_size
        self.dtype = dtype
        self.lock = threading.Lock()

    def _bcast_ctrl(self, *vals):
        ctrl = torch.zeros(_CTRL_LEN, dtype=torch.int64, device=self.device)
        for i, v in enumerate(vals):
            ctrl[i] = int(v)
        dist.broadcast(ctrl, src=0, group=self.group)

    def _bcast_tensor(self, t: torch.Tensor):
        dist.broadcast(t.to(self.device, self.dtype).contiguous(), src=0, group=self.group)

    def span(self, start: int, end: int):
        with self.lock:
            self._bcast_ctrl(OP_SPAN, start, end)

    def open_session(self, slot: int, batch: int, max_length: int):
        with self.lock:
            self._bcast_ctrl(OP_OPEN, slot, batch, max_length)

    def close_session(self, slot: int):
        with self.lock:
            self._bcast_ctrl(OP_CLOSE, slot)

    def _bcast_prompts(self, prompts):
        """prompts: [n_blocks, b, pre_seq, H] or None — deep-ptune prompts are
        added between blocks, so shadows need them to keep activations equal."""
        if prompts is None:
            return
        dist.broadcast(prompts.to(self.device, self.dtype).contiguous(), src=0, group=self.group)

    @staticmethod
    def _prompt_dims(prompts):
        return (1, prompts.shape[0], prompts.shape[2]) if prompts is not None else (0, 0, 0)

    def step(self, slot: int, h: torch.Tensor, position: int, hypo_ids=None, max_chunk: int = 0,
             prompts=None):
        with self.lock:
            b, n = h.shape[0], h.shape[1]
            hp, _nb, pre = self._prompt_dims(prompts)
            self._bcast_ctrl(OP_STEP, slot, b, n, position, 1 if hypo_ids is not None else 0,
                             max_chunk, hp * pre)
            self._bcast_tensor(h.view(b, n, self.hidden_size))
            if hypo_ids is not None:
                hy = hypo_ids.to(self.device, torch.int64).contiguous()
                dist.broadcast(hy, src=0, group=self.group)
            self._bcast_prompts(prompts)

    def forward(self, h: torch.Tensor, prompts=None):
        with self.lock:
            b, s = h.shape[0], h.shape[1]
            hp, _nb, pre = self._prompt_dims(prompts)
            self._bcast_ctrl(OP_FWD, b, s, hp * pre)
            self._bcast_tensor(h)
            self._bcast_prompts(prompts)

    def backward(self, inputs: torch.Tensor, grad_outputs: torch.Tensor, prompts=None):
        with self.lock:
            b, s = inputs.shape[0], inputs.shape[1]
            hp, _nb, pre = self._prompt_dims(prompts)
            self._bcast_ctrl(OP_BWD, b, s, hp * pre)
            self._bcast_tensor(inputs)
            self._bcast_tensor(grad_outputs)
            self._bcast_prompts(prompts)

    def shutdown(self):
        with self.lock:
            try:
                self._bcast_ctrl(OP_SHUTDOWN)
            except Exception:  # noqa: BLE001
                pass


class TPShadowWorker:
    """Ranks 1..N-1: receive ops, run the local shard chain so its collectives
    pair with rank 0's. Holds shard KV caches per (session slot, block)."""

    def __init__(self, model_name_or_dir: str, config, *, device, torch_dtype, quant_type: str,
                 group=None, rank: int, world: int):
        self.model_name_or_dir = model_name_or_dir
        self.config = config
        self.device = torch.device(device)
        self.dtype = torch_dtype
        self.quant = quant_type
        self.group = group
        self.rank, self.world = rank, world
        self.blocks = []
        self.span = (0, 0)
        self.sessions = {}  # slot -> {"caches": [(k, v)...], "position": int}

    def _recv_ctrl(self):
        ctrl = torch.zeros(_CTRL_LEN, dtype=torch.int64, device=self.device)
        dist.broadcast(ctrl, src=0, group=self.group)
        return [int(x) for x in ctrl.tolist()]

    def _recv_tensor(self, *shape):
        t = torch.empty(*shape, device=self.device, dtype=self.dtype)
        dist.broadcast(t, src=0, group=self.group)
        return t

    def _load_span(self, start: int, end: int):
        from petals_amd.server.from_pretrained import load_pretrained_block

        self.blocks = [
            load_pretrained_block(
                self.model_name_or_dir, self.config, i, torch_dtype=self.dtype,
                device=self.device, quant_type=self.quant,
                tp_rank=self.rank, tp_world=self.world, tp_group=self.group,
            )
            for i in range(start, end)
        ]
        self.span = (start, end)
        self.sessions.clear()

    def serve_forever(self):
        H = self.config.hidden_size
        while True:
            op, a, b, c, d, e, f, *_ = self._recv_ctrl()
            if op == OP_SHUTDOWN:
                return
            if op == OP_SPAN:
                self._load_span(a, b)
            elif op == OP_OPEN:
                caches = []
                for blk in self.blocks:
                    ks, vs = blk.kv_cache_shape(b, c)
                    caches.append((
                        torch.zeros(ks, device=self.device, dtype=self.dtype),
                        torch.zeros(vs, device=self.device, dtype=self.dtype),
                    ))
                self.sessions[a] = {"caches": caches}
            elif op == OP_CLOSE:
                self.sessions.pop(a, None)
            elif op == OP_STEP:
                slot, batch, n, position, has_hypo, max_chunk, pre = a, b, c, d, e, f, _[0]
                h = self._recv_tensor(batch, n, H)
                hypo = None
                if has_hypo:
                    hypo = torch.zeros(batch, dtype=torch.int64, device=self.device)
                    dist.broadcast(hypo, src=0, group=self.group)
                prompts = self._recv_prompts(pre, batch, H)
                sess = self.sessions.get(slot)
                if sess is None:
                    continue
                mc = max_chunk if max_chunk > 0 else n
                with torch.inference_mode():
                    if hypo is not None:
                        for k, v in sess["caches"]:
                            k[...] = k[hypo]
                            v[...] = v[hypo]
                    for i, (blk, (k, v)) in enumerate(zip(self.blocks, sess["caches"])):
                        if prompts is not None:
                            h = h.clone()
                            h[:, : prompts.shape[2]] += prompts[i]
                        # chunk boundaries mirror rank0's backend.inference_step
                        # so the collective sequences pair up exactly
                        if n <= mc:
                            h = blk(h, kv_cache=(k, v), prefix_length=position)
                        else:
                            out = torch.empty_like(h)
                            for off in range(0, n, mc):
                                chunk = h[:, off : off + mc]
                                out[:, off : off + chunk.shape[1]] = blk(
                                    chunk, kv_cache=(k, v), prefix_length=position + off
                                )
                            h = out
            elif op == OP_FWD:
                h = self._recv_tensor(a, b, H)
                prompts = self._recv_prompts(c, a, H)
                with torch.inference_mode():
                    for i, blk in enumerate(self.blocks):
                        if prompts is not None:
                            h = h.clone()
                            h[:, : prompts.shape[2]] += prompts[i]
                        h = blk(h)
            elif op == OP_BWD:
                inputs = self._recv_tensor(a, b, H)
                grad_outputs = self._recv_tensor(a, b, H)
                prompts = self._recv_prompts(c, a, H)
                self._run_backward(inputs, grad_outputs, prompts)

    def _recv_prompts(self, pre: int, batch: int, H: int):
        if pre <= 0:
            return None
        p = torch.empty(len(self.blocks), batch, pre, H, device=self.device, dtype=self.dtype)
        dist.broadcast(p, src=0, group=self.group)
        return p

    def _run_backward(self, inputs, grad_outputs, prompts=None):
        """Mirror handler._backward_chain: no-grad forward to recover
        intermediate inputs, then per-block autograd backward in reverse —
        same collective order as rank 0."""
        inter = []
        hidden = inputs

        def add_prompt(h, i):
            if prompts is None:
                return h
            h = h.clone()
            h[:, : prompts.shape[2]] += prompts[i]
            return h

        with torch.no_grad():
            for i, blk in enumerate(self.blocks[:-1]):
                hidden = add_prompt(hidden, i)
                inter.append(hidden)
                hidden = blk(hidden)
            inter.append(add_prompt(hidden, len(self.blocks) - 1))
        grad = grad_outputs
        for blk, hid in zip(reversed(self.blocks), reversed(inter)):
            with torch.enable_grad():
                x = hid.detach().requires_grad_(True)
                out = blk(x)
                out.backward(grad)
                grad = x.grad
