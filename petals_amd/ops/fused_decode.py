"""MI355X fast path for Llama-family blocks.

`LlamaFastPath` repacks a loaded LlamaBlock's weights into kernel-optimal
layout (transposed [in, out] bf16, QKV and gate/up concatenated) and then runs:

  decode (q_len == 1, any batch <= 8):
      rms_norm_f32out -> gemv(QKV) -> rope_cache_write -> attn_decode_fused
      -> gemv(O, +residual) -> rms_norm_f32out -> gemv(gate|up, SwiGLU epi)
      -> gemv(down, +residual)
    ... 8 kernel launches per block, every byte of weights read exactly once
    (HBM-bound by design; see profiles/).

  prefill (q_len > 1): bf16 GEMMs via torch.matmul on the SAME transposed
    weights (rocBLAS), HIP rms_norm / rope / swiglu kernels, fp32-softmax
    attention (chunked by the backend; flash prefill kernel is the planned
    upgrade).

The original nn.Linear weights are FREED after repacking (a 70B span must not
hold two copies).
"""

from __future__ import annotations

import math
import os
from typing import Dict, Optional, Tuple

import torch

from petals_amd.ops import reference

_EPI_RAW = -1  # return split partials; caller fuses its own reduce+epilogue
_EPI_PLAIN_F32 = 0
_EPI_PLAIN_BF16 = 1
_EPI_RESIDUAL_BF16 = 2
_EPI_SWIGLU_F32 = 3
_EPI_GELU_F32 = 4

_workspaces: Dict[Tuple[torch.device, str], torch.Tensor] = {}
_retired_workspaces: list = []  # replaced but possibly graph-captured (see _get_ws)


def _get_ws(device: torch.device, key: str, numel: int) -> torch.Tensor:
    k = (device, key)
    ws = _workspaces.get(k)
    if ws is None or ws.numel() < numel:
        if ws is not None:
            # NEVER free a replaced workspace: hipGraphs capture raw pointers,
            # so freeing it lets the allocator hand the memory to someone else
            # and a previously captured graph replays into foreign storage.
            # (Root cause of the round-1 "bloom dual-graph hardware fault":
            # the client LM-head graph for bloom's 250k vocab grew the shared
            # gemv workspace AFTER the span graph had captured the old one.)
            _retired_workspaces.append(ws)
        ws = torch.empty(numel, dtype=torch.float32, device=device)
        _workspaces[k] = ws
    return ws


class DecodeContext:
    """Device-resident decode position shared by all blocks of a step.

    Makes the whole token step hipGraph-capturable: kernels read the position
    from device memory, so a captured graph replays correctly as `advance()`
    (or `set_position`) moves the position — no host-side re-capture."""

    norm_parts = None  # folded-RMSNorm: producer-side sum(h^2) partials handed
    # from block n-1's down-projection reduce to block n's qkv gemv; span
    # loops MUST reset this to None before their first block each pass

    def __init__(self, device: torch.device):
        self.pos = torch.zeros(1, dtype=torch.int32, device=device)  # prefix length
        self.kv_len = torch.zeros(1, dtype=torch.int32, device=device)  # prefix + 1

    def set_position(self, prefix_length: int) -> None:
        self.pos.fill_(prefix_length)
        self.kv_len.fill_(prefix_length + 1)

    def advance(self) -> None:
        """On-device increment (graph-capturable)."""
        self.pos.add_(1)
        self.kv_len.add_(1)


_split_autotune_cache: Dict[Tuple[str, int, int], int] = {}


class _FastWeight:
    """A transposed [in, out] weight in bf16, NF4 or weight-only int8
    (quantize-on-load; int8 = per-out-column symmetric absmax, the MI355X
    stand-in for the reference's optional LLM.int8 path).

    Split-K counts are autotuned once per (kind, in, out) shape at load time
    (profiles/gemv_split_sweep.log shows the optimum varies ~2x by shape);
    the result is cached process-wide so only the first block pays."""

    def __init__(self, t_bf16: torch.Tensor, hip, quant: str):
        self.hip = hip
        self.quant = quant
        self.in_dim, self.out_dim = t_bf16.shape
        self.t = self.packed = self.absmax = self.absmax_t = self.q8 = self.scale8 = None
        if quant == "nf4":
            self.packed, self.absmax = hip.nf4_quantize(t_bf16.contiguous())
            # transposed absmax copy: one contiguous 32 B load per 16-row
            # unroll block in the gemv instead of 16 strided 2 B gathers
            self.absmax_t = self.absmax.t().contiguous()
        elif quant == "int8":
            w = t_bf16.float()
            scale = w.abs().amax(dim=0).clamp_min(1e-8) / 127.0
            self.q8 = torch.round(w / scale).clamp(-127, 127).to(torch.int8).contiguous()
            self.scale8 = scale.to(torch.bfloat16).contiguous()
        else:
            self.t = t_bf16.contiguous()
        self.splits = self._autotune(t_bf16.device)

    def _autotune(self, device) -> int:
        key = (self.quant, self.in_dim, self.out_dim)
        if key in _split_autotune_cache:
            return _split_autotune_cache[key]
        if device.type != "cuda":
            return 0
        import time

        if self.quant == "nf4":
            candidates = [0, 32, 64, 96, 128, 160, 192, 256, 320, 384]
            max_chunk = 32
        elif self.quant == "int8":
            candidates = [0, 16, 32, 64, 128, 192]
            max_chunk = 128
        else:
            candidates = [0, 8, 16, 32, 64, 128]
            max_chunk = 64
        candidates = [s for s in candidates if s == 0 or s * max_chunk <= self.in_dim]
        x = torch.randn(1, self.in_dim, device=device)
        ws = _get_ws(device, "gemv", 64 * max(self.out_dim, 1))
        best, best_t = 0, float("inf")
        for s in candidates:
            try:
                self._gemv_raw(x, ws, None, 0, s)  # warm
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                for _ in range(8):
                    self._gemv_raw(x, ws, None, 0, s)
                torch.cuda.synchronize()
                dt = time.perf_counter() - t0
            except RuntimeError:
                continue
            if dt < best_t:
                best, best_t = s, dt
        _split_autotune_cache[key] = best
        return best

    @property
    def shape(self):
        return (self.in_dim, self.out_dim)

    def _gemv_raw(self, x, ws, residual, epilogue, splits, bias=None, x_parts=None,
                  fold_eps=0.0, sumsq_out=None):
        if self.quant == "nf4":
            return self.hip.gemv_nf4(self.packed, self.absmax, x, ws, residual, epilogue, splits, bias,
                                     self.absmax_t, x_parts, fold_eps, sumsq_out)
        assert x_parts is None and sumsq_out is None, "folded-norm gemv is NF4-only"
        if self.quant == "int8":
            return self.hip.gemv_int8(self.q8, self.scale8, x, ws, residual, epilogue, splits, bias)
        return self.hip.gemv_bf16(self.t, x, ws, residual, epilogue, splits, bias)

    def gemv(self, x, ws, residual, epilogue, bias=None, x_parts=None, fold_eps=0.0, sumsq_out=None):
        return self._gemv_raw(x, ws, residual, epilogue, self.splits, bias,
                              x_parts=x_parts, fold_eps=fold_eps, sumsq_out=sumsq_out)

    def dense(self) -> torch.Tensor:
        """bf16 [in, out] view for prefill GEMMs. Quantized weights dequantize
        through a process-wide LRU (default 16 GiB of the 288 GB HBM,
        PETALS_AMD_DEQUANT_CACHE_MB) so chunked prefills and back-to-back
        sessions stop paying the per-matmul dequant (~337 us/weight measured
        round 1, PARITY.md)."""
        if self.quant == "nf4":
            return _dequant_cached(self, lambda: self.hip.nf4_dequantize(self.packed, self.absmax))
        if self.quant == "int8":
            return _dequant_cached(
                self, lambda: (self.q8.to(torch.float32) * self.scale8.float()).to(torch.bfloat16)
            )
        return self.t


# ---- dequantized-weight LRU (keyed by _FastWeight identity, byte-bounded)
_dequant_cache: "OrderedDict[int, torch.Tensor]" = None  # type: ignore[assignment]
_dequant_cache_bytes = 0


def _dequant_cache_budget() -> int:
    import os as _os

    return int(_os.environ.get("PETALS_AMD_DEQUANT_CACHE_MB", "16384")) << 20


def _dequant_cached(w, make) -> torch.Tensor:
    global _dequant_cache, _dequant_cache_bytes
    from collections import OrderedDict

    if _dequant_cache is None:
        _dequant_cache = OrderedDict()
    key = id(w)
    t = _dequant_cache.get(key)
    if t is not None:
        _dequant_cache.move_to_end(key)
        return t
    t = make()
    budget = _dequant_cache_budget()
    nbytes = t.numel() * t.element_size()
    if nbytes > budget:
        return t  # too big to cache at all
    while _dequant_cache and _dequant_cache_bytes + nbytes > budget:
        _k, old = _dequant_cache.popitem(last=False)
        _dequant_cache_bytes -= old.numel() * old.element_size()
    _dequant_cache[key] = t
    _dequant_cache_bytes += nbytes
    import weakref

    def _evict(k=key):
        global _dequant_cache_bytes
        old = _dequant_cache.pop(k, None) if _dequant_cache else None
        if old is not None:
            _dequant_cache_bytes -= old.numel() * old.element_size()

    weakref.finalize(w, _evict)  # id() reuse after GC must not alias entries
    return t


def _nf4_decode_max_b() -> int:
    # native BATCH 5-8 kernels exist (UNROLL drops 16->8 to fit the VGPR
    # budget); sub-batching at 4 vs native-8 is an A/B — the measured winner
    # is the default (profiles/bench_serve_batched.log)
    return int(os.environ.get("PETALS_AMD_NF4_DECODE_MAXB", "8"))


def decode_step_auto(fast, hidden, k_cache, v_cache, prefix_length: int = -1, ctx=None, adapter=None,
                     max_b: int = None):
    """Fused decode for any batch <= 8: batches past the kernel cap split
    into sub-batches — the weights are re-read once per sub-batch, which
    still beats falling back to the dense (dequantizing) prefill path by
    ~5x at batch 8."""
    if max_b is None:
        max_b = _nf4_decode_max_b() if fast.quant == "nf4" else 8
    B = hidden.shape[0]
    if B <= max_b:
        return fast.decode_step(hidden, k_cache, v_cache, prefix_length, ctx=ctx, adapter=adapter)
    outs = []
    for i in range(0, B, max_b):
        outs.append(
            fast.decode_step(
                hidden[i : i + max_b], k_cache[i : i + max_b], v_cache[i : i + max_b],
                prefix_length, ctx=ctx, adapter=adapter,
            )
        )
    return torch.cat(outs, dim=0)


class _TPFastPathMixin:
    """Tensor-parallel hooks shared by the per-family fast paths. The shard's
    row-parallel outputs (o/down/dense) skip the fused residual epilogue and
    are all-reduced before the residual add (parallel/tp.py)."""

    tp_world = 1
    tp_group = None

    def _tp_copy(self, x: torch.Tensor) -> torch.Tensor:
        """Column-parallel input boundary (identity fwd, grad all-reduce bwd)."""
        if self.tp_world <= 1:
            return x
        from petals_amd.parallel.tp import copy_to_tp

        return copy_to_tp(x, self.tp_group)

    def _tp_allreduce_(self, t: torch.Tensor) -> None:
        """In-place SUM across TP ranks (decode hot path; overridable in tests)."""
        import torch.distributed as dist

        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self.tp_group)

    def _tp_reduce(self, x: torch.Tensor) -> torch.Tensor:
        """Sum row-parallel partials across TP ranks (autograd-aware)."""
        if self.tp_world <= 1:
            return x
        from petals_amd.parallel.tp import reduce_from_tp

        return reduce_from_tp(x, self.tp_group)


class LlamaFastPath(_TPFastPathMixin):
    graph_safe = True  # whole-span hipGraph capture is valid for this block

    def __init__(self, block, hip_ops, quant: str = "none", tp_world: int = 1, tp_group=None):
        cfg = block.config
        self.hip = hip_ops
        self.cfg = cfg
        self.quant = quant
        # tensor parallelism (parallel/tp.py): cfg already IS the shard
        # geometry; row-parallel outputs (o/down) skip the fused residual
        # epilogue and are all-reduced before the residual add
        self.tp_world = tp_world
        self.tp_group = tp_group
        if tp_world > 1:
            # RCCL collectives inside hipGraph capture are not validated yet
            self.graph_safe = False
        self.hd = cfg.head_dim
        self.qh = cfg.num_attention_heads
        self.kh = cfg.n_kv_heads
        self.gq = self.qh // self.kh
        self.scale = 1.0 / math.sqrt(self.hd)
        self.eps = cfg.layer_norm_eps

        attn = block.self_attn
        device = attn.q_proj.weight.device

        def t(w):
            return w.detach().to(torch.bfloat16).t().contiguous()

        # folded RMSNorm (RMSNorm(x)·W == (x·inv_rms) @ diag(ln_w)·W): the norm
        # WEIGHT is pre-multiplied into the quantized weight rows, and decode
        # derives inv_rms from producer-side sum(h^2) partials — removing both
        # norm kernels per block from the decode chain. MEASURED NEGATIVE end
        # to end (74.2 vs 76.3 tok/s, profiles/fold_norm_ab.log): the sumsq
        # emission + per-wave accumulator-scale tails across ~10k gemv waves
        # cost more than the two removed 4.8 us norm kernels, even with the
        # parts loads prefetched before the weight stream. Default OFF;
        # PETALS_AMD_FOLD_NORM=1 keeps the path selectable (numerics covered
        # by test_folded_norm_decode_matches_unfolded).
        self.fold_norm = (
            quant == "nf4"
            and tp_world == 1
            and getattr(type(self), "supports_fold_norm", True)
            and cfg.hidden_size % 64 == 0
            and cfg.hidden_size // 64 <= 128  # XS gemv prefetches <= 128 partials
            and os.environ.get("PETALS_AMD_FOLD_NORM", "0") == "1"
        )
        self.ln1_w = block.input_layernorm.weight.detach().to(torch.bfloat16).contiguous()
        self.ln2_w = block.post_attention_layernorm.weight.detach().to(torch.bfloat16).contiguous()
        if self.fold_norm:
            ln1f = block.input_layernorm.weight.detach().float().to(device)

            def t1(w):
                return (w.detach().float().t().to(device) * ln1f[:, None]).to(torch.bfloat16).contiguous()
        else:
            t1 = t

        self.wqkv_t = _FastWeight(
            torch.cat([t1(attn.q_proj.weight), t1(attn.k_proj.weight), t1(attn.v_proj.weight)], dim=1),
            hip_ops, quant,
        )  # [H, qh*hd + 2*kh*hd]
        self.wo_t = _FastWeight(t(attn.o_proj.weight), hip_ops, quant)
        # norm weights as seen by the NON-fold code paths (prefill, autograd,
        # adapters): identity when folded into the matmul weights
        if self.fold_norm:
            self.ln1_w_post = torch.ones_like(self.ln1_w)
            self.ln2_w_post = torch.ones_like(self.ln2_w)
            P = cfg.hidden_size // 64
            self._ss1 = torch.empty(8, P, dtype=torch.float32, device=device)
            self._ss2 = torch.empty(8, P, dtype=torch.float32, device=device)
        else:
            self.ln1_w_post = self.ln1_w
            self.ln2_w_post = self.ln2_w
        self._empty_bf16 = torch.empty(0, device=device, dtype=torch.bfloat16)
        for lin in (attn.q_proj, attn.k_proj, attn.v_proj, attn.o_proj):
            lin.weight.data = self._empty_bf16
        self.has_bias = attn.q_proj.bias is not None
        self._init_mlp_weights(block, hip_ops, quant)

        self.rope_cos: Optional[torch.Tensor] = None
        self.rope_sin: Optional[torch.Tensor] = None
        self._pos = torch.zeros(1, dtype=torch.int32, device=device)
        self._kv_len = torch.zeros(1, dtype=torch.int32, device=device)
        self.device = device
        self._empty_f32 = torch.empty(0, dtype=torch.float32, device=device)

    def _init_mlp_weights(self, block, hip_ops, quant):
        def t(w):
            return w.detach().to(torch.bfloat16).t().contiguous()

        if self.fold_norm:
            ln2f = block.post_attention_layernorm.weight.detach().float().to(self.ln2_w.device)

            def t2(w):
                return (w.detach().float().t().to(ln2f.device) * ln2f[:, None]).to(torch.bfloat16).contiguous()
        else:
            t2 = t

        mlp = block.mlp
        self.wgateup_t = _FastWeight(
            torch.cat([t2(mlp.gate_proj.weight), t2(mlp.up_proj.weight)], dim=1), hip_ops, quant
        )
        self.wdown_t = _FastWeight(t(mlp.down_proj.weight), hip_ops, quant)
        for lin in (mlp.gate_proj, mlp.up_proj, mlp.down_proj):
            lin.weight.data = self._empty_bf16

    def _mlp_dense(self, xn2, adapter, autograd: bool, x_delta=None):
        """MLP on [B, S, H] inputs via rocBLAS matmuls on the transposed
        (possibly NF4-dequantized) weights; differentiable when autograd.
        `x_delta`: adapter-delta input when it differs from the matmul input
        (folded-norm weights take the scale-only normed x)."""
        xn2 = self._tp_copy(xn2)
        xd = xn2 if x_delta is None else x_delta
        gateup = torch.matmul(xn2, self.wgateup_t.dense())
        inter = self.wgateup_t.shape[1] // 2
        gate, up = gateup[..., :inter], gateup[..., inter:]
        if adapter is not None:
            dg, du = adapter.delta("gate", xd), adapter.delta("up", xd)
            if dg is not None:
                gate = gate + dg
            if du is not None:
                up = up + du
        if autograd:
            act = reference.swiglu(gate, up).to(xn2.dtype)
        else:
            act = self.hip.swiglu(gate.contiguous(), up.contiguous())
        down = torch.matmul(act, self.wdown_t.dense())
        if adapter is not None:
            dd = adapter.delta("down", act)
            if dd is not None:
                down = down + dd
        return down

    def _ensure_rope(self, needed: int):
        if self.rope_cos is None or self.rope_cos.shape[0] < needed:
            cos, sin = reference.build_rope_cache(
                self.hd,
                max(needed, self.cfg.max_position_embeddings),
                theta=self.cfg.rope_theta,
                rope_scaling=self.cfg.rope_scaling,
            )
            self.rope_cos = cos.to(self.device, torch.float32).contiguous()
            self.rope_sin = sin.to(self.device, torch.float32).contiguous()

    # ------------------------------------------------------------- decode

    @torch.inference_mode()
    def decode_step(
        self,
        hidden: torch.Tensor,  # [B, 1, H] bf16
        k_cache: torch.Tensor,  # [Bc, KV, Lmax, hd] bf16
        v_cache: torch.Tensor,
        prefix_length: int = -1,
        ctx: Optional[DecodeContext] = None,
        adapter=None,
    ) -> torch.Tensor:
        B = hidden.shape[0]
        H = hidden.shape[-1]
        h = hidden.view(B, H)
        if h.dtype != torch.bfloat16:
            h = h.to(torch.bfloat16)
        h = h.contiguous()
        if ctx is None:
            assert prefix_length >= 0
            self._ensure_rope(prefix_length + 1)
            self._pos.fill_(prefix_length)
            self._kv_len.fill_(prefix_length + 1)
            pos, kv_len = self._pos, self._kv_len
        else:
            # position lives on device (graph-capturable); rope table must
            # already cover the cache (ensure_rope(lmax) at session start)
            self._ensure_rope(k_cache.shape[2])
            pos, kv_len = ctx.pos, ctx.kv_len
        ws = _get_ws(self.device, "gemv", 64 * B * self._max_gemv_out())

        fold = self.fold_norm and adapter is None
        if fold:
            # folded RMSNorm: no norm kernels at all — the qkv gemv consumes the
            # raw bf16 hidden plus producer-side sum(h^2) partials (from the
            # previous block's down-reduce via ctx, or one sumsq kernel at the
            # span head); the mlp gemv consumes the o-reduce's partials
            parts_in = getattr(ctx, "norm_parts", None) if ctx is not None else None
            if parts_in is None:
                parts_in = self.hip.sumsq_rows(h)
            parts = self.wqkv_t.gemv(h, ws, None, _EPI_RAW, x_parts=parts_in, fold_eps=self.eps)
            q = self.hip.qkv_rope_reduce(
                parts, self.rope_cos, self.rope_sin, pos, k_cache[:B], v_cache[:B],
                self.qh, self.kh, True,
            )
        elif adapter is None:
            xn = self.hip.rms_norm_f32out(h, self.ln1_w_post, self.eps)  # [B, H] f32
            # fused qkv reduce + rope + cache write: one kernel fewer per block
            parts = self.wqkv_t.gemv(xn, ws, None, _EPI_RAW)
            q = self.hip.qkv_rope_reduce(
                parts, self.rope_cos, self.rope_sin, pos, k_cache[:B], v_cache[:B],
                self.qh, self.kh, True,
            )  # [B, qh*hd] f32, rotated
        else:
            xn = self.hip.rms_norm_f32out(h, self.ln1_w_post, self.eps)
            qkv = self.wqkv_t.gemv(xn, ws, None, _EPI_PLAIN_F32)  # [B, qkv] f32
            # adapters were trained against the true normed activations: undo
            # the fold for the delta input only
            xn_a = xn * self.ln1_w.float() if self.fold_norm else xn
            self._apply_qkv_adapter(qkv, xn_a, adapter)
            self.hip.rope_cache_write(
                qkv, self.rope_cos, self.rope_sin, pos, k_cache[:B], v_cache[:B], self.qh, self.kh
            )
            q = qkv[:, : self.qh * self.hd].contiguous()
        attn = self.hip.attn_decode_fused(
            q, k_cache[:B], v_cache[:B], kv_len, self.gq, 0,
            self._empty_f32, self._empty_f32, self.scale,
        )  # [B, qh*hd] f32 (the shard's heads under TP)
        if self.tp_world > 1:
            part = self.wo_t.gemv(attn, ws, None, _EPI_PLAIN_F32)  # [B, H] f32 partial
            self._tp_allreduce_(part)
            h2 = (h.float() + part).to(torch.bfloat16)
        elif adapter is not None:
            d = adapter.delta("o", attn)
            h2 = self.wo_t.gemv(attn, ws, h, _EPI_RESIDUAL_BF16)  # [B, H] bf16
            if d is not None:
                h2 = h2 + d.to(h2.dtype)
        elif fold:
            h2 = self.wo_t.gemv(attn, ws, h, _EPI_RESIDUAL_BF16, sumsq_out=self._ss1[:B])
            act = self.wgateup_t.gemv(h2, ws, None, _EPI_SWIGLU_F32,
                                      x_parts=self._ss1[:B], fold_eps=self.eps)
            h3 = self.wdown_t.gemv(act, ws, h2, _EPI_RESIDUAL_BF16, sumsq_out=self._ss2[:B])
            if ctx is not None:
                ctx.norm_parts = self._ss2[:B]
            return h3.view(B, 1, H)
        else:
            h2 = self.wo_t.gemv(attn, ws, h, _EPI_RESIDUAL_BF16)  # [B, H] bf16
        xn2 = self.hip.rms_norm_f32out(h2, self.ln2_w_post, self.eps)
        h3 = self._mlp_decode(xn2, h2, ws, adapter)
        return h3.view(B, 1, H)

    def _max_gemv_out(self) -> int:
        return max(self.wgateup_t.shape[1], self.wqkv_t.shape[1], self.wo_t.shape[1])

    def _mlp_decode(self, xn2, h2, ws, adapter):
        if self.tp_world > 1:
            assert adapter is None, "LoRA on TP shards is served via the generic path"
            act = self.wgateup_t.gemv(xn2, ws, None, _EPI_SWIGLU_F32)  # [B, I/world] f32
            part = self.wdown_t.gemv(act, ws, None, _EPI_PLAIN_F32)  # [B, H] f32 partial
            self._tp_allreduce_(part)
            return (h2.float() + part).to(torch.bfloat16)
        if adapter is None:
            act = self.wgateup_t.gemv(xn2, ws, None, _EPI_SWIGLU_F32)  # [B, I] f32
            return self.wdown_t.gemv(act, ws, h2, _EPI_RESIDUAL_BF16)
        gateup = self.wgateup_t.gemv(xn2, ws, None, _EPI_PLAIN_F32)  # [B, 2I] f32
        inter = self.wgateup_t.shape[1] // 2
        xd = xn2 * self.ln2_w.float() if self.fold_norm else xn2  # true normed x for deltas
        dg, du = adapter.delta("gate", xd), adapter.delta("up", xd)
        gate, up = gateup[:, :inter], gateup[:, inter:]
        if dg is not None:
            gate = gate + dg.float()
        if du is not None:
            up = up + du.float()
        act = (torch.nn.functional.silu(gate) * up).contiguous()
        h3 = self.wdown_t.gemv(act, ws, h2, _EPI_RESIDUAL_BF16)
        dd = adapter.delta("down", act)
        if dd is not None:
            h3 = h3 + dd.to(h3.dtype)
        return h3

    # --------------------------------------------------- training (autograd)

    def _qkv_adapter_delta(self, x, adapter):
        """Returns the concatenated qkv LoRA delta for inputs [..., H]."""
        qd, kd = self.qh * self.hd, self.kh * self.hd
        parts = []
        for key, width in (("q", qd), ("k", kd), ("v", kd)):
            d = adapter.delta(key, x)
            parts.append(d if d is not None else x.new_zeros(*x.shape[:-1], width))
        return torch.cat(parts, dim=-1)

    def _apply_qkv_adapter(self, qkv, x, adapter):
        """Adds LoRA deltas into the fused qkv buffer (decode fast path)."""
        qd = self.qh * self.hd
        kd = self.kh * self.hd
        for key, lo, hi in (("q", 0, qd), ("k", qd, qd + kd), ("v", qd + kd, qd + 2 * kd)):
            d = adapter.delta(key, x)
            if d is not None:
                qkv[:, lo:hi] += d.float()

    def forward_autograd(self, hidden: torch.Tensor, prefix_length: int = 0, adapter=None) -> torch.Tensor:
        """Differentiable forward on the transposed weights (torch primitives
        only — the HIP kernels are inference-only). Used by rpc_backward on GPU
        servers; weights are frozen, grads flow to inputs/prompts."""
        B, S, H = hidden.shape
        end = prefix_length + S
        self._ensure_rope(end)
        xn = reference.rms_norm(hidden, self.ln1_w_post, self.eps)
        xn = self._tp_copy(xn)  # backward: all-reduce the shard-partial grads
        qkv = torch.matmul(xn, self.wqkv_t.dense())
        if adapter is not None:
            xd = reference.rms_norm(hidden, self.ln1_w, self.eps) if self.fold_norm else xn
            qkv = qkv + self._qkv_adapter_delta(xd, adapter)
        q = qkv[..., : self.qh * self.hd].view(B, S, self.qh, self.hd).transpose(1, 2)
        k = qkv[..., self.qh * self.hd : (self.qh + self.kh) * self.hd].view(B, S, self.kh, self.hd).transpose(1, 2)
        v = qkv[..., (self.qh + self.kh) * self.hd :].view(B, S, self.kh, self.hd).transpose(1, 2)
        pos = torch.arange(prefix_length, end, device=hidden.device)
        q, k = reference.apply_rope(q, k, self.rope_cos, self.rope_sin, pos)
        attn = reference.attention(q, k, v, causal=True)
        attn = attn.transpose(1, 2).reshape(B, S, self.qh * self.hd).to(hidden.dtype)
        o = self._tp_reduce(torch.matmul(attn, self.wo_t.dense()))
        if adapter is not None:
            d = adapter.delta("o", attn)
            if d is not None:
                o = o + d
        h2 = hidden + o
        xn2 = reference.rms_norm(h2, self.ln2_w_post, self.eps)
        xd2 = reference.rms_norm(h2, self.ln2_w, self.eps) if (self.fold_norm and adapter is not None) else None
        return h2 + self._tp_reduce(self._mlp_dense(xn2, adapter, autograd=True, x_delta=xd2)).to(h2.dtype)

    # ------------------------------------------------------------ prefill

    def forward(
        self,
        hidden: torch.Tensor,  # [B, S, H] bf16
        kv_cache: Optional[Tuple[torch.Tensor, torch.Tensor]],
        prefix_length: int,
        adapter=None,
    ) -> torch.Tensor:
        B, S, H = hidden.shape
        hidden = hidden.to(torch.bfloat16)
        end = prefix_length + S
        self._ensure_rope(end)

        xn = self.hip.rms_norm(hidden, self.ln1_w_post, self.eps)
        qkv = torch.matmul(xn, self.wqkv_t.dense())  # [B, S, qkv] bf16 (rocBLAS)
        if adapter is not None:
            xd = self.hip.rms_norm(hidden, self.ln1_w, self.eps) if self.fold_norm else xn
            qkv = qkv + self._qkv_adapter_delta(xd, adapter).to(qkv.dtype)
        q = qkv[..., : self.qh * self.hd].view(B, S, self.qh, self.hd).transpose(1, 2)
        k = qkv[..., self.qh * self.hd : (self.qh + self.kh) * self.hd].view(B, S, self.kh, self.hd).transpose(1, 2)
        v = qkv[..., (self.qh + self.kh) * self.hd :].view(B, S, self.kh, self.hd).transpose(1, 2)
        pos = torch.arange(prefix_length, end, device=self.device).unsqueeze(0).expand(B, S).contiguous()
        q, k = self.hip.apply_rope(q.contiguous(), k.contiguous(), self.rope_cos, self.rope_sin, pos)
        if kv_cache is not None:
            k_cache, v_cache = kv_cache
            k_cache[:B, :, prefix_length:end].copy_(k)
            v_cache[:B, :, prefix_length:end].copy_(v)
            attn = self.hip.attn_prefill_fused(
                q.contiguous(), k_cache[:B].contiguous(), v_cache[:B].contiguous(),
                end, prefix_length, self.scale, True,
            )
        else:
            attn = self.hip.attn_prefill_fused(
                q.contiguous(), k.contiguous(), v.contiguous(), S, 0, self.scale, True
            )
        attn = attn.transpose(1, 2).reshape(B, S, self.qh * self.hd).to(torch.bfloat16)
        o = self._tp_reduce(torch.matmul(attn, self.wo_t.dense()))
        if adapter is not None:
            d = adapter.delta("o", attn)
            if d is not None:
                o = o + d
        h2 = hidden + o
        xn2 = self.hip.rms_norm(h2, self.ln2_w_post, self.eps)
        xd2 = self.hip.rms_norm(h2, self.ln2_w, self.eps) if (self.fold_norm and adapter is not None) else None
        return h2 + self._tp_reduce(self._mlp_dense(xn2, adapter, autograd=False, x_delta=xd2)).to(h2.dtype)


# ---------------------------------------------------------------------------
# BLOOM: LayerNorm -> fused-QKV (per-head interleaved in the checkpoint,
# re-permuted to [q|k|v] at load) -> ALiBi MHA -> LayerNorm -> GELU MLP,
# biases on every projection. Decode runs the same 8-kernel chain as Llama
# with layer_norm/kv_cache_write/alibi variants (all graph-capturable).
# Parity: reference models/bloom/block.py wraps HF BloomBlock in torch.
# ---------------------------------------------------------------------------


def _bloom_qkv_perm(qh: int, hd: int) -> torch.Tensor:
    """new [q|k|v] column index -> old per-head-interleaved [h,3,hd] index."""
    idx = torch.arange(qh * hd)
    heads, d = idx // hd, idx % hd
    perm = torch.empty(3 * qh * hd, dtype=torch.long)
    for which in range(3):
        perm[which * qh * hd + idx] = heads * 3 * hd + which * hd + d
    return perm


class BloomFastPath(_TPFastPathMixin):
    # The round-1 "bloom dual-graph hardware fault" (a bloom span graph +
    # the client's LM-head graph faulting on replay while each was clean in
    # isolation) was ROOT-CAUSED to the shared gemv workspace: bloom's 250k
    # vocab made the head's workspace request GROW the shared buffer, freeing
    # the allocation the span graph had captured raw pointers into. _get_ws
    # now retires (never frees) replaced workspaces, and the dual-graph
    # pattern is exercised by tests/test_gpu_kernels.py::
    # test_bloom_dual_graph_with_big_vocab_head.
    graph_safe = True

    def __init__(self, block, hip_ops, quant: str = "none", tp_world: int = 1, tp_group=None):
        cfg = block.config
        self.hip = hip_ops
        self.cfg = cfg
        self.quant = quant
        self.tp_world = tp_world
        self.tp_group = tp_group
        if tp_world > 1:
            self.graph_safe = False  # RCCL inside hipGraph capture unvalidated
        self.hd = cfg.head_dim
        self.qh = cfg.num_attention_heads
        self.kh = self.qh  # MHA
        self.gq = 1
        self.scale = 1.0 / math.sqrt(self.hd)
        self.eps = cfg.layer_norm_eps
        self.post_ln_residual = bool(getattr(cfg, "apply_residual_connection_post_layernorm", False))

        attn = block.self_attention
        device = attn.query_key_value.weight.device

        def t(w):
            return w.detach().to(torch.bfloat16).t().contiguous()

        def bias(lin):
            return lin.bias.detach().to(torch.bfloat16).contiguous()

        perm = _bloom_qkv_perm(self.qh, self.hd).to(device)
        self._qkv_perm = perm  # HF->flat [q|k|v] column map (LoRA deltas reuse it)
        self.wqkv_t = _FastWeight(t(attn.query_key_value.weight)[:, perm].contiguous(), hip_ops, quant)
        self.qkv_bias = attn.query_key_value.bias.detach().to(torch.bfloat16)[perm].contiguous()
        self.wo_t = _FastWeight(t(attn.dense.weight), hip_ops, quant)
        self.o_bias = bias(attn.dense)
        mlp = block.mlp
        self.w_h4h = _FastWeight(t(mlp.dense_h_to_4h.weight), hip_ops, quant)
        self.b_h4h = bias(mlp.dense_h_to_4h)
        self.w_4hh = _FastWeight(t(mlp.dense_4h_to_h.weight), hip_ops, quant)
        self.b_4hh = bias(mlp.dense_4h_to_h)
        self.ln1_w = block.input_layernorm.weight.detach().to(torch.bfloat16).contiguous()
        self.ln1_b = block.input_layernorm.bias.detach().to(torch.bfloat16).contiguous()
        self.ln2_w = block.post_attention_layernorm.weight.detach().to(torch.bfloat16).contiguous()
        self.ln2_b = block.post_attention_layernorm.bias.detach().to(torch.bfloat16).contiguous()
        self._empty_bf16 = torch.empty(0, device=device, dtype=torch.bfloat16)
        for lin in (attn.query_key_value, attn.dense, mlp.dense_h_to_4h, mlp.dense_4h_to_h):
            lin.weight.data = self._empty_bf16

        # ALiBi slopes depend on the GLOBAL head index: a TP shard takes its
        # slice of the full model's slope table
        total = getattr(cfg, "alibi_total_heads", None) or self.qh
        start = getattr(cfg, "alibi_start_head", 0)
        self.slopes = (
            reference.build_alibi_slopes(total)[start : start + self.qh]
            .to(device=device, dtype=torch.float32)
            .contiguous()
        )
        self._pos = torch.zeros(1, dtype=torch.int32, device=device)
        self._kv_len = torch.zeros(1, dtype=torch.int32, device=device)
        self.device = device
        self._empty_f32 = torch.empty(0, dtype=torch.float32, device=device)

    def _max_gemv_out(self) -> int:
        return max(self.wqkv_t.shape[1], self.w_h4h.shape[1])

    # ------------------------------------------------------------- decode

    @torch.inference_mode()
    def decode_step(self, hidden, k_cache, v_cache, prefix_length: int = -1, ctx=None, adapter=None):
        B, H = hidden.shape[0], hidden.shape[-1]
        h = hidden.view(B, H)
        if h.dtype != torch.bfloat16:
            h = h.to(torch.bfloat16)
        h = h.contiguous()
        if ctx is None:
            assert prefix_length >= 0
            self._pos.fill_(prefix_length)
            self._kv_len.fill_(prefix_length + 1)
            pos, kv_len = self._pos, self._kv_len
        else:
            pos, kv_len = ctx.pos, ctx.kv_len
        ws = _get_ws(self.device, "gemv", 64 * B * self._max_gemv_out())

        xn = self.hip.layer_norm_f32out(h, self.ln1_w, self.ln1_b, self.eps)
        res1 = self.hip.layer_norm(h, self.ln1_w, self.ln1_b, self.eps) if self.post_ln_residual else h
        if adapter is None:
            parts = self.wqkv_t.gemv(xn, ws, None, _EPI_RAW)
            q = self.hip.qkv_rope_reduce(
                parts, None, None, pos, k_cache[:B], v_cache[:B], self.qh, self.kh, False,
                bias=self.qkv_bias,
            )  # ALiBi family: reduce + bias + cache write, no rotation
        else:
            assert self.tp_world == 1, "LoRA on TP shards is served via the generic path"
            qkv = self.wqkv_t.gemv(xn, ws, None, _EPI_PLAIN_F32, bias=self.qkv_bias)
            d = adapter.delta("qkv", xn)
            if d is not None:
                qkv += d.float()[:, self._qkv_perm]  # deltas arrive in the HF column order
            self.hip.kv_cache_write(qkv, pos, k_cache[:B], v_cache[:B], self.qh, self.kh)
            q = qkv[:, : self.qh * self.hd].contiguous()
        attn = self.hip.attn_decode_fused(
            q, k_cache[:B], v_cache[:B], kv_len, self.gq, 0,
            self._empty_f32, self._empty_f32, self.scale, self.slopes,
        )
        if self.tp_world > 1:
            # row-parallel dense: partial (rank-0-only bias baked in by the TP
            # shard loader), all-reduce, then the replicated residual add
            part = self.wo_t.gemv(attn, ws, None, _EPI_PLAIN_F32, bias=self.o_bias)
            self._tp_allreduce_(part)
            h2 = (res1.float() + part).to(torch.bfloat16)
        else:
            h2 = self.wo_t.gemv(attn, ws, res1, _EPI_RESIDUAL_BF16, bias=self.o_bias)
            if adapter is not None:
                dd = adapter.delta("dense", attn)
                if dd is not None:
                    h2 = h2 + dd.to(h2.dtype)
        xn2 = self.hip.layer_norm_f32out(h2, self.ln2_w, self.ln2_b, self.eps)
        res2 = self.hip.layer_norm(h2, self.ln2_w, self.ln2_b, self.eps) if self.post_ln_residual else h2
        if adapter is not None:
            inter = self.w_h4h.gemv(xn2, ws, None, _EPI_PLAIN_F32, bias=self.b_h4h)
            dh = adapter.delta("h4h", xn2)
            if dh is not None:
                inter += dh.float()
            act = reference.gelu(inter).contiguous()
        else:
            act = self.w_h4h.gemv(xn2, ws, None, _EPI_GELU_F32, bias=self.b_h4h)
        if self.tp_world > 1:
            part = self.w_4hh.gemv(act, ws, None, _EPI_PLAIN_F32, bias=self.b_4hh)
            self._tp_allreduce_(part)
            h3 = (res2.float() + part).to(torch.bfloat16)
        else:
            h3 = self.w_4hh.gemv(act, ws, res2, _EPI_RESIDUAL_BF16, bias=self.b_4hh)
            if adapter is not None:
                d4 = adapter.delta("4hh", act)
                if d4 is not None:
                    h3 = h3 + d4.to(h3.dtype)
        return h3.view(B, 1, H)

    # -------------------------------------------------- prefill / training

    def _split_heads(self, qkv, B, S):
        qd = self.qh * self.hd
        q = qkv[..., :qd].view(B, S, self.qh, self.hd).transpose(1, 2)
        k = qkv[..., qd : 2 * qd].view(B, S, self.kh, self.hd).transpose(1, 2)
        v = qkv[..., 2 * qd :].view(B, S, self.kh, self.hd).transpose(1, 2)
        return q, k, v

    def _body(self, hidden, kv_cache, prefix_length, autograd: bool, adapter=None):
        B, S, H = hidden.shape
        hidden = hidden.to(torch.bfloat16)
        end = prefix_length + S
        fln = torch.nn.functional.layer_norm

        def ln(x, w, b):
            if autograd or x.device.type != "cuda":
                return fln(x.float(), (H,), w.float(), b.float(), self.eps).to(torch.bfloat16)
            return self.hip.layer_norm(x, w, b, self.eps)

        xn = ln(hidden, self.ln1_w, self.ln1_b)
        res1 = xn if self.post_ln_residual else hidden
        xn = self._tp_copy(xn)
        qkv = torch.matmul(xn, self.wqkv_t.dense()) + self.qkv_bias
        if adapter is not None:
            d = adapter.delta("qkv", xn)
            if d is not None:
                qkv = qkv + d.to(qkv.dtype)[..., self._qkv_perm]
        q, k, v = self._split_heads(qkv, B, S)
        if kv_cache is not None and not autograd:
            k_cache, v_cache = kv_cache
            k_cache[:B, :, prefix_length:end].copy_(k)
            v_cache[:B, :, prefix_length:end].copy_(v)
            attn = self.hip.attn_prefill_fused(
                q.contiguous(), k_cache[:B].contiguous(), v_cache[:B].contiguous(),
                end, prefix_length, self.scale, True, self.slopes,
            )
        elif not autograd and hidden.device.type == "cuda":
            attn = self.hip.attn_prefill_fused(
                q.contiguous(), k.contiguous(), v.contiguous(), S, 0, self.scale, True, self.slopes
            )
        else:
            k_pos = torch.arange(end, device=hidden.device, dtype=torch.float32)
            bias = (self.slopes.to(hidden.device)[:, None, None] * k_pos[None, None, :]).unsqueeze(0)
            attn = reference.attention(q.float(), k.float(), v.float(), causal=True,
                                       kv_offset=prefix_length, attn_bias=bias.float())
        attn = attn.transpose(1, 2).reshape(B, S, self.qh * self.hd).to(torch.bfloat16)
        o = torch.matmul(attn, self.wo_t.dense()) + self.o_bias
        if adapter is not None:
            d = adapter.delta("dense", attn)
            if d is not None:
                o = o + d.to(o.dtype)
        h2 = res1 + self._tp_reduce(o)
        xn2 = ln(h2, self.ln2_w, self.ln2_b)
        res2 = xn2 if self.post_ln_residual else h2
        xn2 = self._tp_copy(xn2)
        inter = torch.matmul(xn2, self.w_h4h.dense()) + self.b_h4h
        if adapter is not None:
            dh = adapter.delta("h4h", xn2)
            if dh is not None:
                inter = inter + dh.to(inter.dtype)
        act = reference.gelu(inter.float()).to(torch.bfloat16)
        out = torch.matmul(act, self.w_4hh.dense()) + self.b_4hh
        if adapter is not None:
            d4 = adapter.delta("4hh", act)
            if d4 is not None:
                out = out + d4.to(out.dtype)
        return res2 + self._tp_reduce(out)

    def forward(self, hidden, kv_cache, prefix_length, adapter=None):
        assert adapter is None or self.tp_world == 1, "LoRA on TP shards is served via the generic path"
        return self._body(hidden, kv_cache, prefix_length, autograd=False, adapter=adapter)

    def forward_autograd(self, hidden, prefix_length: int = 0, adapter=None):
        assert adapter is None or self.tp_world == 1, "LoRA on TP shards is served via the generic path"
        return self._body(hidden, None, prefix_length, autograd=True, adapter=adapter)


# ---------------------------------------------------------------------------
# Falcon (new_decoder_architecture: 40B/180B): parallel attention + MLP with
# two LayerNorms on the same input, rope GQA, no biases, GELU MLP.
# Decode: ln_attn/ln_mlp -> qkv gemv -> rope+cache write -> GQA flash decode
# -> dense gemv (+resid) -> gelu gemv -> down gemv (+resid) = 9 kernels.
# Parity: reference models/falcon/block.py wraps HF FalconDecoderLayer and
# CUDA-graphs only the QKV split.
# ---------------------------------------------------------------------------


def _falcon_qkv_perm(qh: int, kh: int, hd: int) -> torch.Tensor:
    """new [q|k|v] column -> old [(kv group)(gq q heads, k, v)(hd)] column."""
    gq = qh // kh
    group = (gq + 2) * hd
    perm = torch.empty((qh + 2 * kh) * hd, dtype=torch.long)
    iq = torch.arange(qh * hd)
    kvg, rem = (iq // hd) // gq, iq % (hd)
    slot = (iq // hd) % gq
    perm[iq] = kvg * group + slot * hd + rem
    ik = torch.arange(kh * hd)
    perm[qh * hd + ik] = (ik // hd) * group + gq * hd + (ik % hd)
    perm[(qh + kh) * hd + ik] = (ik // hd) * group + (gq + 1) * hd + (ik % hd)
    return perm


class FalconFastPath(_TPFastPathMixin):
    graph_safe = True

    def __init__(self, block, hip_ops, quant: str = "none", tp_world: int = 1, tp_group=None):
        cfg = block.config
        self.tp_world = tp_world
        self.tp_group = tp_group
        if tp_world > 1:
            self.graph_safe = False  # RCCL inside hipGraph capture unvalidated
        # two fused geometries: the new-decoder architecture (40B/180B:
        # ln_attn + ln_mlp, GQA) and the 7B-style old decoder (single
        # input_layernorm feeding BOTH attn and mlp, parallel residual, MQA —
        # its fused-QKV layout equals the new-decoder layout with n_kv=1)
        self.single_ln = not cfg.new_decoder_architecture
        if self.single_ln:
            assert cfg.parallel_attn, "fused Falcon old-decoder path needs parallel_attn"
        self.hip = hip_ops
        self.cfg = cfg
        self.quant = quant
        self.hd = cfg.head_dim
        self.qh = cfg.num_attention_heads
        self.kh = cfg.n_kv_heads
        self.gq = self.qh // self.kh
        self.scale = 1.0 / math.sqrt(self.hd)
        self.eps = cfg.layer_norm_eps

        attn = block.self_attention
        device = attn.query_key_value.weight.device

        def t(w):
            return w.detach().to(torch.bfloat16).t().contiguous()

        perm = _falcon_qkv_perm(self.qh, self.kh, self.hd).to(device)
        self._qkv_perm = perm  # HF->flat [q|k|v] column map (LoRA deltas reuse it)
        self.wqkv_t = _FastWeight(t(attn.query_key_value.weight)[:, perm].contiguous(), hip_ops, quant)
        self.wo_t = _FastWeight(t(attn.dense.weight), hip_ops, quant)
        mlp = block.mlp
        self.w_h4h = _FastWeight(t(mlp.dense_h_to_4h.weight), hip_ops, quant)
        self.w_4hh = _FastWeight(t(mlp.dense_4h_to_h.weight), hip_ops, quant)
        ln_attn = block.input_layernorm if self.single_ln else block.ln_attn
        ln_mlp = block.input_layernorm if self.single_ln else block.ln_mlp
        self.ln_attn_w = ln_attn.weight.detach().to(torch.bfloat16).contiguous()
        self.ln_attn_b = ln_attn.bias.detach().to(torch.bfloat16).contiguous()
        self.ln_mlp_w = ln_mlp.weight.detach().to(torch.bfloat16).contiguous()
        self.ln_mlp_b = ln_mlp.bias.detach().to(torch.bfloat16).contiguous()
        self._empty_bf16 = torch.empty(0, device=device, dtype=torch.bfloat16)
        for lin in (attn.query_key_value, attn.dense, mlp.dense_h_to_4h, mlp.dense_4h_to_h):
            lin.weight.data = self._empty_bf16

        self.rope_cos: Optional[torch.Tensor] = None
        self.rope_sin: Optional[torch.Tensor] = None
        self._pos = torch.zeros(1, dtype=torch.int32, device=device)
        self._kv_len = torch.zeros(1, dtype=torch.int32, device=device)
        self.device = device
        self._empty_f32 = torch.empty(0, dtype=torch.float32, device=device)

    def _ensure_rope(self, needed: int):
        if self.rope_cos is None or self.rope_cos.shape[0] < needed:
            cos, sin = reference.build_rope_cache(
                self.hd, max(needed, self.cfg.max_position_embeddings), theta=self.cfg.rope_theta
            )
            self.rope_cos = cos.to(self.device, torch.float32).contiguous()
            self.rope_sin = sin.to(self.device, torch.float32).contiguous()

    def _max_gemv_out(self) -> int:
        return max(self.wqkv_t.shape[1], self.w_h4h.shape[1])

    @torch.inference_mode()
    def decode_step(self, hidden, k_cache, v_cache, prefix_length: int = -1, ctx=None, adapter=None):
        B, H = hidden.shape[0], hidden.shape[-1]
        h = hidden.view(B, H)
        if h.dtype != torch.bfloat16:
            h = h.to(torch.bfloat16)
        h = h.contiguous()
        if ctx is None:
            assert prefix_length >= 0
            self._ensure_rope(prefix_length + 1)
            self._pos.fill_(prefix_length)
            self._kv_len.fill_(prefix_length + 1)
            pos, kv_len = self._pos, self._kv_len
        else:
            self._ensure_rope(k_cache.shape[2])
            pos, kv_len = ctx.pos, ctx.kv_len
        ws = _get_ws(self.device, "gemv", 64 * B * self._max_gemv_out())

        xn_attn = self.hip.layer_norm_f32out(h, self.ln_attn_w, self.ln_attn_b, self.eps)
        xn_mlp = xn_attn if self.single_ln else self.hip.layer_norm_f32out(h, self.ln_mlp_w, self.ln_mlp_b, self.eps)
        if adapter is None:
            parts = self.wqkv_t.gemv(xn_attn, ws, None, _EPI_RAW)
            q = self.hip.qkv_rope_reduce(
                parts, self.rope_cos, self.rope_sin, pos, k_cache[:B], v_cache[:B],
                self.qh, self.kh, True,
            )
        else:
            assert self.tp_world == 1, "LoRA on TP shards is served via the generic path"
            qkv = self.wqkv_t.gemv(xn_attn, ws, None, _EPI_PLAIN_F32)  # flat [q|k|v] f32
            d = adapter.delta("qkv", xn_attn)
            if d is not None:
                qkv += d.float()[:, self._qkv_perm]  # deltas arrive in the HF column order
            self.hip.rope_cache_write(
                qkv, self.rope_cos, self.rope_sin, pos, k_cache[:B], v_cache[:B], self.qh, self.kh
            )
            q = qkv[:, : self.qh * self.hd].contiguous()
        attn = self.hip.attn_decode_fused(
            q, k_cache[:B], v_cache[:B], kv_len, self.gq, 0,
            self._empty_f32, self._empty_f32, self.scale,
        )
        if adapter is not None:
            h2 = self.wo_t.gemv(attn, ws, h, _EPI_RESIDUAL_BF16)
            dd = adapter.delta("dense", attn)
            if dd is not None:
                h2 = h2 + dd.to(h2.dtype)
            inter = self.w_h4h.gemv(xn_mlp, ws, None, _EPI_PLAIN_F32)
            dh = adapter.delta("h4h", xn_mlp)
            if dh is not None:
                inter += dh.float()
            act = reference.gelu(inter).contiguous()
            h3 = self.w_4hh.gemv(act, ws, h2, _EPI_RESIDUAL_BF16)
            d4 = adapter.delta("4hh", act)
            if d4 is not None:
                h3 = h3 + d4.to(h3.dtype)
            return h3.view(B, 1, H)
        if self.tp_world > 1:
            # parallel residual (attn + mlp both row-parallel partials):
            # sum the two partials locally, then ONE all-reduce per block
            ap = self.wo_t.gemv(attn, ws, None, _EPI_PLAIN_F32)
            act = self.w_h4h.gemv(xn_mlp, ws, None, _EPI_GELU_F32)
            part = self.w_4hh.gemv(act, ws, None, _EPI_PLAIN_F32)
            part.add_(ap)
            self._tp_allreduce_(part)
            h3 = (h.float() + part).to(torch.bfloat16)
            return h3.view(B, 1, H)
        h2 = self.wo_t.gemv(attn, ws, h, _EPI_RESIDUAL_BF16)  # resid + attn
        act = self.w_h4h.gemv(xn_mlp, ws, None, _EPI_GELU_F32)
        h3 = self.w_4hh.gemv(act, ws, h2, _EPI_RESIDUAL_BF16)  # ... + mlp
        return h3.view(B, 1, H)

    def _split_heads(self, qkv, B, S):
        qd, kd = self.qh * self.hd, self.kh * self.hd
        q = qkv[..., :qd].view(B, S, self.qh, self.hd).transpose(1, 2)
        k = qkv[..., qd : qd + kd].view(B, S, self.kh, self.hd).transpose(1, 2)
        v = qkv[..., qd + kd :].view(B, S, self.kh, self.hd).transpose(1, 2)
        return q, k, v

    def _body(self, hidden, kv_cache, prefix_length, autograd: bool, adapter=None):
        B, S, H = hidden.shape
        hidden = hidden.to(torch.bfloat16)
        end = prefix_length + S
        self._ensure_rope(end)
        fln = torch.nn.functional.layer_norm

        def ln(x, w, b):
            if autograd or x.device.type != "cuda":
                return fln(x.float(), (H,), w.float(), b.float(), self.eps).to(torch.bfloat16)
            return self.hip.layer_norm(x, w, b, self.eps)

        xn_attn = self._tp_copy(ln(hidden, self.ln_attn_w, self.ln_attn_b))
        xn_mlp = xn_attn if self.single_ln else self._tp_copy(ln(hidden, self.ln_mlp_w, self.ln_mlp_b))
        qkv = torch.matmul(xn_attn, self.wqkv_t.dense())
        if adapter is not None:
            d = adapter.delta("qkv", xn_attn)
            if d is not None:
                qkv = qkv + d.to(qkv.dtype)[..., self._qkv_perm]
        q, k, v = self._split_heads(qkv, B, S)
        pos = torch.arange(prefix_length, end, device=hidden.device)
        if autograd:
            q, k = reference.apply_rope(q, k, self.rope_cos, self.rope_sin, pos)
            attn = reference.attention(q.float(), k.float(), v.float(), causal=True, kv_offset=prefix_length)
        else:
            posb = pos.unsqueeze(0).expand(B, S).contiguous()
            q, k = self.hip.apply_rope(q.contiguous(), k.contiguous(), self.rope_cos, self.rope_sin, posb)
            if kv_cache is not None:
                k_cache, v_cache = kv_cache
                k_cache[:B, :, prefix_length:end].copy_(k)
                v_cache[:B, :, prefix_length:end].copy_(v)
                attn = self.hip.attn_prefill_fused(
                    q.contiguous(), k_cache[:B].contiguous(), v_cache[:B].contiguous(),
                    end, prefix_length, self.scale, True,
                )
            else:
                attn = self.hip.attn_prefill_fused(
                    q.contiguous(), k.contiguous(), v.contiguous(), S, 0, self.scale, True
                )
        attn = attn.transpose(1, 2).reshape(B, S, self.qh * self.hd).to(torch.bfloat16)
        o = torch.matmul(attn, self.wo_t.dense())
        inter = torch.matmul(xn_mlp, self.w_h4h.dense())
        if adapter is not None:
            d = adapter.delta("dense", attn)
            if d is not None:
                o = o + d.to(o.dtype)
            dh = adapter.delta("h4h", xn_mlp)
            if dh is not None:
                inter = inter + dh.to(inter.dtype)
        act = reference.gelu(inter.float()).to(torch.bfloat16)
        mlp_out = torch.matmul(act, self.w_4hh.dense())
        if adapter is not None:
            d4 = adapter.delta("4hh", act)
            if d4 is not None:
                mlp_out = mlp_out + d4.to(mlp_out.dtype)
        return hidden + self._tp_reduce(o + mlp_out)

    def forward(self, hidden, kv_cache, prefix_length, adapter=None):
        assert adapter is None or self.tp_world == 1, "LoRA on TP shards is served via the generic path"
        return self._body(hidden, kv_cache, prefix_length, autograd=False, adapter=adapter)

    def forward_autograd(self, hidden, prefix_length: int = 0, adapter=None):
        assert adapter is None or self.tp_world == 1, "LoRA on TP shards is served via the generic path"
        return self._body(hidden, None, prefix_length, autograd=True, adapter=adapter)
