"""MI355X fast path for Mixtral blocks: fused Llama-style attention decode +
device-routed expert GEMVs.

Decode routing runs entirely ON the GPU (router gemm -> softmax -> top-k ->
int32 expert ids in a device tensor) and the expert GEMVs read their expert's
stacked weights through that tensor (ops/csrc/moe.hip), so there is no host
sync per step and the whole span is hipGraph-capturable (graph_safe) for the
bf16 and NF4 quantizations. Each selected expert's weights are read exactly
once per token — the decode-time degenerate case of a grouped GEMM.

(The reference runs Mixtral experts densely inside one server's HF block with
host-side routing: reference models/mixtral/block.py:73-81.)

int8 expert weights keep the round-1 host-routed per-expert GEMV path
(graph_safe = False).
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn.functional as F

from petals_amd.ops.fused_decode import (
    _EPI_PLAIN_F32,
    _EPI_SWIGLU_F32,
    LlamaFastPath,
    _FastWeight,
    _dequant_cached,
)


MOE_GEMM_MT = 32  # moe_gemm C-tile rows (ops/csrc/moe.hip MOE_MT)


def sort_pairs_by_expert(selected: torch.Tensor, num_experts: int):
    """Expert-sorted (token, slot) pair ids with each expert's segment padded
    to MT rows, so every MT-row GEMM tile belongs to one expert (all device
    ops — no host sync; the pad buffer is sized by the static upper bound).
    Returns (sorted_pairs [Rp] i32 with -1 padding, tile_expert [Rp/MT] i32)."""
    MT = MOE_GEMM_MT
    dev = selected.device
    sel = selected.reshape(-1)  # [R] expert per pair
    R = sel.shape[0]
    order = torch.argsort(sel, stable=True)
    sel_sorted = sel[order]
    counts = torch.bincount(sel, minlength=num_experts)
    padded = (counts + MT - 1) // MT * MT
    pad_off = torch.cumsum(padded, 0) - padded       # padded segment starts
    unpad_off = torch.cumsum(counts, 0) - counts     # unpadded segment starts
    slot = pad_off[sel_sorted] + (torch.arange(R, device=dev) - unpad_off[sel_sorted])
    rp = (R + num_experts * (MT - 1) + MT - 1) // MT * MT  # static upper bound
    sorted_pairs = torch.full((rp,), -1, dtype=torch.int32, device=dev)
    sorted_pairs[slot] = order.to(torch.int32)
    tile_expert = torch.full((rp // MT,), -1, dtype=torch.int32, device=dev)
    tile_expert[slot // MT] = sel_sorted.to(torch.int32)
    return sorted_pairs, tile_expert


class _CacheKey:
    """Weakref-able identity key for the dequant LRU (plain object() cannot
    carry weak references, which _dequant_cached uses for eviction)."""

    __slots__ = ("__weakref__",)


class _StackedExperts:
    """All experts' transposed [in, out] weights stacked on a leading expert
    axis, in the exact layout the moe.hip kernels stream."""

    def __init__(self, t_list: List[torch.Tensor], hip, quant: str):
        self.hip = hip
        self.quant = quant
        self.in_dim, self.out_dim = t_list[0].shape
        if quant == "nf4":
            packed, absmax = [], []
            for t in t_list:
                p, a = hip.nf4_quantize(t.contiguous())
                packed.append(p)
                absmax.append(a)
            self.packed_all = torch.stack(packed).contiguous()
            self.absmax_all = torch.stack(absmax).contiguous()
        else:
            assert quant == "none"
            self.wt_all = torch.stack([t.contiguous() for t in t_list]).contiguous()
        # per-expert identity keys for the dequant LRU (fused_decode._dequant_cached
        # keys by object id): multi-chunk prefills reuse each expert's dense form
        self._cache_keys = [_CacheKey() for _ in t_list]

    def moe_gemv(self, x: torch.Tensor, sel: torch.Tensor, k_per_tok: int, ws: torch.Tensor,
                 epilogue: int) -> torch.Tensor:
        if self.quant == "nf4":
            return self.hip.gemv_nf4_moe(self.packed_all, self.absmax_all, x, sel, k_per_tok, ws, epilogue)
        return self.hip.gemv_bf16_moe(self.wt_all, x, sel, k_per_tok, ws, epilogue)

    def moe_gemm(self, x: torch.Tensor, sorted_pairs: torch.Tensor, tile_expert: torch.Tensor,
                 k_per_tok: int, n_rows: int) -> torch.Tensor:
        """Grouped MFMA GEMM (prefill): C[pair] = x[pair // k] @ W[expert(pair)]."""
        if self.quant == "nf4":
            return self.hip.moe_gemm(None, self.packed_all, self.absmax_all, x,
                                     sorted_pairs, tile_expert, k_per_tok, n_rows)
        return self.hip.moe_gemm(self.wt_all, None, None, x, sorted_pairs, tile_expert,
                                 k_per_tok, n_rows)

    @property
    def gemm_ok(self) -> bool:
        return self.out_dim % 128 == 0 and self.in_dim % 64 == 0

    def dense(self, e: int) -> torch.Tensor:
        """Expert e as a dense bf16 [in, out] (prefill/training matmuls),
        LRU-cached so chunked prefills pay the dequant once per expert."""
        if self.quant == "nf4":
            return _dequant_cached(
                self._cache_keys[e],
                lambda: self.hip.nf4_dequantize(self.packed_all[e], self.absmax_all[e]),
            )
        return self.wt_all[e]


class MixtralFastPath(LlamaFastPath):
    # the router consumes the true ln2-weighted normed activations (softmax is
    # not scale-invariant), so the MLP norm cannot fold into expert weights
    supports_fold_norm = False

    def _init_mlp_weights(self, block, hip_ops, quant):
        def t(w):
            return w.detach().to(torch.bfloat16).t().contiguous()

        moe = block.block_sparse_moe
        self.top_k = moe.top_k
        self.num_experts = moe.num_experts
        self.w_router_t = t(moe.gate.weight)  # [H, E]
        self.stacked_gu: Optional[_StackedExperts] = None
        self.stacked_down: Optional[_StackedExperts] = None
        self.wgateup_e: List[_FastWeight] = []
        self.wdown_e: List[_FastWeight] = []
        if quant in ("none", "nf4"):
            gu_list = [torch.cat([t(e.w1.weight), t(e.w3.weight)], dim=1) for e in moe.experts]
            down_list = [t(e.w2.weight) for e in moe.experts]
            self.stacked_gu = _StackedExperts(gu_list, hip_ops, quant)
            self.stacked_down = _StackedExperts(down_list, hip_ops, quant)
            self.graph_safe = True  # device-routed decode: capturable
        else:
            for expert in moe.experts:
                self.wgateup_e.append(
                    _FastWeight(torch.cat([t(expert.w1.weight), t(expert.w3.weight)], dim=1), hip_ops, quant)
                )
                self.wdown_e.append(_FastWeight(t(expert.w2.weight), hip_ops, quant))
            self.graph_safe = False  # host-routed fallback
        for expert in moe.experts:
            for lin in (expert.w1, expert.w2, expert.w3):
                lin.weight.data = self._empty_bf16
        moe.gate.weight.data = self._empty_bf16

    def _max_gemv_out(self) -> int:
        gu_out = self.stacked_gu.out_dim if self.stacked_gu is not None else self.wgateup_e[0].shape[1]
        return max(gu_out, self.wqkv_t.shape[1], self.wo_t.shape[1])

    def _route(self, xn2: torch.Tensor):
        """Router on [tokens, H] f32/bf16 -> (weights [t, k], experts [t, k]).
        All-device: safe inside hipGraph capture."""
        logits = xn2.to(self.w_router_t.dtype) @ self.w_router_t
        probs = F.softmax(logits.float(), dim=-1)
        weights, selected = torch.topk(probs, self.top_k, dim=-1)
        weights = weights / weights.sum(dim=-1, keepdim=True)
        return weights, selected

    def _mlp_decode(self, xn2, h2, ws, adapter):
        """xn2 f32 [B, H]; returns h3 bf16 [B, H]."""
        B = xn2.shape[0]
        weights, selected = self._route(xn2)
        if self.stacked_gu is not None:
            sel = selected.to(torch.int32).reshape(-1).contiguous()  # [B*k]
            act = self.stacked_gu.moe_gemv(xn2.contiguous(), sel, self.top_k, ws, _EPI_SWIGLU_F32)
            down = self.stacked_down.moe_gemv(act.contiguous(), sel, 1, ws, _EPI_PLAIN_F32)
            moe_out = (down.view(B, self.top_k, -1) * weights.unsqueeze(-1)).sum(dim=1)
            return (h2.float() + moe_out).to(torch.bfloat16)

        # int8: host-synced routing (~10 us, amortized against MBs of expert reads)
        sel_l = selected.tolist()
        w_l = weights.tolist()
        moe_out = torch.zeros_like(xn2)  # f32 [B, H]
        for b in range(B):
            xb = xn2[b : b + 1].contiguous()
            for j in range(self.top_k):
                e = sel_l[b][j]
                act = self.wgateup_e[e].gemv(xb, ws, None, _EPI_SWIGLU_F32)  # [1, I] f32
                down = self.wdown_e[e].gemv(act, ws, None, _EPI_PLAIN_F32)  # [1, H] f32
                moe_out[b] += down[0] * w_l[b][j]
        return (h2.float() + moe_out).to(torch.bfloat16)

    def _expert_dense(self, e: int) -> torch.Tensor:
        return self.stacked_gu.dense(e) if self.stacked_gu is not None else self.wgateup_e[e].dense()

    def _expert_down_dense(self, e: int) -> torch.Tensor:
        return self.stacked_down.dense(e) if self.stacked_down is not None else self.wdown_e[e].dense()

    def _mlp_dense(self, xn2, adapter, autograd: bool, x_delta=None):
        """Prefill / training MLP over the routed experts. Inference prefill
        runs the grouped MFMA GEMM (ops/csrc/moe.hip moe_gemm: NF4 dequant
        fused into the LDS B-tile staging — packed weights are what cross
        HBM, read ceil(rows_e/32) times per expert instead of once per
        token); training/CPU fall back to token-grouped dense matmuls."""
        shape = xn2.shape
        x = xn2.reshape(-1, shape[-1])
        weights, selected = self._route(x)
        if (
            not autograd
            and x.device.type == "cuda"
            and self.stacked_gu is not None
            and self.stacked_gu.gemm_ok
            and self.stacked_down.gemm_ok
        ):
            T, K = x.shape[0], self.top_k
            sorted_pairs, tile_expert = sort_pairs_by_expert(selected, self.num_experts)
            xb = x.to(torch.bfloat16).contiguous()
            gateup = self.stacked_gu.moe_gemm(xb, sorted_pairs, tile_expert, K, T * K)
            inter = gateup.shape[-1] // 2
            act = self.hip.swiglu(gateup[:, :inter].contiguous(), gateup[:, inter:].contiguous())
            down = self.stacked_down.moe_gemm(act.contiguous(), sorted_pairs, tile_expert, 1, T * K)
            out = (down.view(T, K, -1).float() * weights.unsqueeze(-1)).sum(dim=1)
            return out.reshape(*shape[:-1], -1).to(xn2.dtype)
        out_dim = self.stacked_down.out_dim if self.stacked_down is not None else self.wdown_e[0].shape[1]
        out = torch.zeros(x.shape[0], out_dim, dtype=torch.float32, device=x.device)
        expert_mask = F.one_hot(selected, num_classes=self.num_experts).permute(2, 1, 0)
        for e in range(self.num_experts):
            k_idx, tok_idx = torch.where(expert_mask[e])
            if tok_idx.numel() == 0:
                continue
            xe = x[tok_idx].to(torch.bfloat16)
            gateup = xe @ self._expert_dense(e)
            inter = gateup.shape[-1] // 2
            act = F.silu(gateup[..., :inter].float()) * gateup[..., inter:].float()
            down = act.to(torch.bfloat16) @ self._expert_down_dense(e)
            out.index_add_(0, tok_idx, down.float() * weights[tok_idx, k_idx, None])
        return out.reshape(*shape[:-1], -1).to(xn2.dtype)
