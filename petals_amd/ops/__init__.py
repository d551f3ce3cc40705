"""Op dispatch: HIP/CDNA4 kernels on MI355X, torch reference on CPU.

Policy: on a GPU box the HIP extension is REQUIRED — ops raise if a CUDA
tensor arrives and `petals_amd._hip_ops` is missing (no silent eager
fallback; set PETALS_AMD_ALLOW_TORCH_FALLBACK=1 to override for debugging).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from petals_amd.ops import reference

_hip_ops = None
_hip_import_error: Optional[BaseException] = None


def _load_hip_ops():
    global _hip_ops, _hip_import_error
    if _hip_ops is not None:
        return _hip_ops
    try:
        from petals_amd.ops import _hip  # thin loader for the in-tree .so

        _hip_ops = _hip.load()
    except BaseException as e:  # noqa: BLE001
        _hip_import_error = e
        _hip_ops = None
    return _hip_ops


def hip_available() -> bool:
    return _load_hip_ops() is not None


def _require_hip(op_name: str):
    ops = _load_hip_ops()
    if ops is None:
        if os.environ.get("PETALS_AMD_ALLOW_TORCH_FALLBACK") == "1":
            return None
        raise RuntimeError(
            f"petals_amd HIP extension is required for {op_name} on GPU but could not be "
            f"loaded (build it with `python -m petals_amd.ops.build` or __graft_entry__.build()). "
            f"Import error: {_hip_import_error!r}"
        )
    return ops


def _grad_mode(*tensors) -> bool:
    """True if this op must stay on differentiable torch ops (training path):
    the HIP kernels are inference-only and would silently break autograd."""
    return torch.is_grad_enabled() and any(t is not None and t.requires_grad for t in tensors)


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda and not _grad_mode(x, weight):
        ops = _require_hip("rms_norm")
        if ops is not None and x.dtype == torch.bfloat16:
            return ops.rms_norm(x.contiguous(), weight.to(torch.bfloat16).contiguous(), eps)
    return reference.rms_norm(x, weight, eps)


def apply_rope(q, k, cos, sin, position_ids):
    if q.is_cuda and not _grad_mode(q, k) and q.dtype == torch.bfloat16:
        ops = _require_hip("apply_rope")
        if ops is not None:
            if position_ids.dim() == 1:
                position_ids = position_ids.unsqueeze(0).expand(q.shape[0], -1)
            out = ops.apply_rope(
                q.contiguous(), k.contiguous(), cos.contiguous(), sin.contiguous(), position_ids.contiguous()
            )
            return out[0], out[1]
    return reference.apply_rope(q, k, cos, sin, position_ids)


def _alibi_bias_matrix(alibi_slopes, kv_len, device):
    """[1, H, 1, kv_len] additive bias = slope * key_pos (fallback paths)."""
    k_pos = torch.arange(kv_len, device=device, dtype=torch.float32)
    return (alibi_slopes.to(device)[:, None, None] * k_pos[None, None, :]).unsqueeze(0)


def attention(q, k, v, *, causal: bool, kv_offset: int = 0, attn_bias=None, scale=None, alibi_slopes=None):
    """Prefill attention. On GPU (bf16, head_dim 64/128, no additive bias
    matrix — ALiBi via per-head `alibi_slopes` stays on the kernel) this runs
    the MFMA flash kernel; otherwise the fp32-softmax torch composition."""
    if (
        q.is_cuda
        and not _grad_mode(q, k, v)
        and attn_bias is None
        and q.dtype == torch.bfloat16
        and k.dtype == torch.bfloat16
        and q.shape[-1] in (64, 128)
        and q.shape[1] % k.shape[1] == 0
    ):
        hops = _require_hip("attention_prefill")
        if hops is not None:
            import math

            kv_len = k.shape[2]
            slopes = None if alibi_slopes is None else alibi_slopes.to(q.device, torch.float32).contiguous()
            return hops.attn_prefill_fused(
                q.contiguous(), k.contiguous(), v.contiguous(), kv_len, int(kv_offset),
                float(scale) if scale is not None else 1.0 / math.sqrt(q.shape[-1]), bool(causal),
                slopes,
            )
    if alibi_slopes is not None and attn_bias is None:
        attn_bias = _alibi_bias_matrix(alibi_slopes, k.shape[2], q.device)
    return reference.attention(q, k, v, causal=causal, kv_offset=kv_offset, attn_bias=attn_bias, scale=scale)


def attention_decode(q, k_cache, v_cache, kv_len: int, *, attn_bias=None, scale=None, alibi_slopes=None):
    """Decode attention over a preallocated cache.

    q: [b, n_heads, q_len, hd]; k_cache/v_cache: [b, n_kv, max_len, hd] with
    valid prefix of length kv_len (incl. the current step already written)."""
    if (
        q.is_cuda
        and not _grad_mode(q)
        and attn_bias is None
        and q.shape[2] == 1
        and q.dtype in (torch.bfloat16, torch.float16)
        and k_cache.dtype == torch.bfloat16
    ):
        ops = _require_hip("attention_decode")
        if ops is not None:
            import math

            b, n_heads, _, hd = q.shape
            n_kv = k_cache.shape[1]
            gq = n_heads // n_kv
            if hd in (64, 128) and gq in (1, 2, 4, 6, 8, 16):
                kv_len_t = torch.tensor([kv_len], dtype=torch.int32, device=q.device)
                empty = torch.empty(0, dtype=torch.float32, device=q.device)
                qf = q.permute(0, 2, 1, 3).reshape(b, n_heads * hd).float().contiguous()
                slopes = None if alibi_slopes is None else alibi_slopes.to(q.device, torch.float32).contiguous()
                out = ops.attn_decode_fused(
                    qf, k_cache.contiguous(), v_cache.contiguous(), kv_len_t, gq, 0, empty, empty,
                    float(scale) if scale is not None else 1.0 / math.sqrt(hd), slopes,
                )
                return out.view(b, 1, n_heads, hd).permute(0, 2, 1, 3).to(q.dtype)
    q_len = q.shape[2]
    if (
        q.is_cuda
        and q_len > 1
        and not _grad_mode(q)
        and attn_bias is None
        and q.dtype == torch.bfloat16
        and k_cache.dtype == torch.bfloat16
        and q.shape[-1] in (64, 128)
        and q.shape[1] % k_cache.shape[1] == 0
    ):
        # multi-token (prefill/chunk) step over the cache: MFMA flash kernel
        hops = _require_hip("attention_prefill")
        if hops is not None:
            import math

            slopes = None if alibi_slopes is None else alibi_slopes.to(q.device, torch.float32).contiguous()
            return hops.attn_prefill_fused(
                q.contiguous(), k_cache.contiguous(), v_cache.contiguous(), int(kv_len),
                int(kv_len) - q_len,
                float(scale) if scale is not None else 1.0 / math.sqrt(q.shape[-1]), True,
                slopes,
            )
    k = k_cache[:, :, :kv_len]
    v = v_cache[:, :, :kv_len]
    if alibi_slopes is not None and attn_bias is None:
        attn_bias = _alibi_bias_matrix(alibi_slopes, kv_len, q.device)
    return reference.attention(
        q, k, v, causal=q_len > 1, kv_offset=kv_len - q_len, attn_bias=attn_bias, scale=scale
    )


def swiglu(gate, up):
    if gate.is_cuda and not _grad_mode(gate, up) and gate.dtype == torch.bfloat16:
        ops = _require_hip("swiglu")
        if ops is not None:
            return ops.swiglu(gate.contiguous(), up.contiguous())
    return reference.swiglu(gate, up)


# re-exports used by blocks
build_rope_cache = reference.build_rope_cache
build_alibi_bias = reference.build_alibi_bias
build_alibi_slopes = reference.build_alibi_slopes
gelu = reference.gelu
repeat_kv = reference.repeat_kv
