"""MI355X fast path for Mixtral blocks: fused Llama-style attention decode +
routed expert GEMVs (decode-time "grouped GEMM" degenerates to top-k indexed
GEMVs on stacked transposed expert weights — each selected expert's weights
are read exactly once per token).

Not hipGraph-safe (expert selection is data-dependent), so the serving span
graph is disabled for Mixtral spans (graph_safe = False); everything else
(8-kernel attention decode, NF4 experts) applies.
"""

from __future__ import annotations

from typing import List

import torch
import torch.nn.functional as F

from petals_amd.ops.fused_decode import (
    _EPI_PLAIN_F32,
    _EPI_SWIGLU_F32,
    LlamaFastPath,
    _FastWeight,
)


class MixtralFastPath(LlamaFastPath):
    graph_safe = False

    def _init_mlp_weights(self, block, hip_ops, quant):
        def t(w):
            return w.detach().to(torch.bfloat16).t().contiguous()

        moe = block.block_sparse_moe
        self.top_k = moe.top_k
        self.num_experts = moe.num_experts
        self.w_router_t = t(moe.gate.weight)  # [H, E]
        self.wgateup_e: List[_FastWeight] = []
        self.wdown_e: List[_FastWeight] = []
        for expert in moe.experts:
            self.wgateup_e.append(
                _FastWeight(torch.cat([t(expert.w1.weight), t(expert.w3.weight)], dim=1), hip_ops, quant)
            )
            self.wdown_e.append(_FastWeight(t(expert.w2.weight), hip_ops, quant))
            for lin in (expert.w1, expert.w2, expert.w3):
                lin.weight.data = self._empty_bf16
        moe.gate.weight.data = self._empty_bf16

    def _max_gemv_out(self) -> int:
        return max(self.wgateup_e[0].shape[1], self.wqkv_t.shape[1], self.wo_t.shape[1])

    def _route(self, xn2: torch.Tensor):
        """Router on [tokens, H] f32/bf16 -> (weights [t, k], experts [t, k])."""
        logits = xn2.to(self.w_router_t.dtype) @ self.w_router_t
        probs = F.softmax(logits.float(), dim=-1)
        weights, selected = torch.topk(probs, self.top_k, dim=-1)
        weights = weights / weights.sum(dim=-1, keepdim=True)
        return weights, selected

    def _mlp_decode(self, xn2, h2, ws, adapter):
        """xn2 f32 [B, H]; returns h3 bf16 [B, H]. Top-k expert GEMVs per row
        (host-synced routing: ~10 us, amortized against ~MBs of expert reads)."""
        B = xn2.shape[0]
        weights, selected = self._route(xn2)
        sel = selected.tolist()
        w = weights.tolist()
        moe_out = torch.zeros_like(xn2)  # f32 [B, H]
        for b in range(B):
            xb = xn2[b : b + 1].contiguous()
            for j in range(self.top_k):
                e = sel[b][j]
                act = self.wgateup_e[e].gemv(xb, ws, None, _EPI_SWIGLU_F32)  # [1, I] f32
                down = self.wdown_e[e].gemv(act, ws, None, _EPI_PLAIN_F32)  # [1, H] f32
                moe_out[b] += down[0] * w[b][j]
        return (h2.float() + moe_out).to(torch.bfloat16)

    def _mlp_dense(self, xn2, adapter, autograd: bool):
        """Prefill / training MLP: token-grouped expert matmuls on the dense
        (dequantized) transposed weights — the grouped-GEMM formulation."""
        shape = xn2.shape
        x = xn2.reshape(-1, shape[-1])
        weights, selected = self._route(x)
        out = torch.zeros(x.shape[0], self.wdown_e[0].shape[1], dtype=torch.float32, device=x.device)
        expert_mask = F.one_hot(selected, num_classes=self.num_experts).permute(2, 1, 0)
        for e in range(self.num_experts):
            k_idx, tok_idx = torch.where(expert_mask[e])
            if tok_idx.numel() == 0:
                continue
            xe = x[tok_idx].to(torch.bfloat16)
            gateup = xe @ self.wgateup_e[e].dense()
            inter = gateup.shape[-1] // 2
            act = F.silu(gateup[..., :inter].float()) * gateup[..., inter:].float()
            down = act.to(torch.bfloat16) @ self.wdown_e[e].dense()
            out.index_add_(0, tok_idx, down.float() * weights[tok_idx, k_idx, None])
        return out.reshape(*shape[:-1], -1).to(xn2.dtype)
