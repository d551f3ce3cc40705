"""Pure-PyTorch reference implementations of every op in the HIP kernel suite.

These are (a) the CPU execution path, and (b) the numerics baseline the HIP
kernels are exact-match tested against (fp32). Keep them obviously correct and
boring; performance lives in `petals_amd/ops/csrc/*.hip`.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    dtype = x.dtype
    x32 = x.to(torch.float32)
    variance = x32.pow(2).mean(-1, keepdim=True)
    x32 = x32 * torch.rsqrt(variance + eps)
    return (x32 * weight.to(torch.float32)).to(dtype)


def build_rope_cache(
    head_dim: int,
    max_len: int,
    theta: float = 10000.0,
    device=None,
    dtype=torch.float32,
    rope_scaling: Optional[dict] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables of shape [max_len, head_dim] (half-dim duplicated, HF
    rotate-half convention). Supports llama3 and linear rope scaling."""
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, dtype=torch.float32) / head_dim))
    if rope_scaling:
        rtype = rope_scaling.get("rope_type", rope_scaling.get("type", ""))
        if rtype == "linear":
            inv_freq = inv_freq / float(rope_scaling["factor"])
        elif rtype == "llama3":
            factor = float(rope_scaling["factor"])
            lo = float(rope_scaling.get("low_freq_factor", 1.0))
            hi = float(rope_scaling.get("high_freq_factor", 4.0))
            orig = float(rope_scaling.get("original_max_position_embeddings", 8192))
            wavelen = 2 * math.pi / inv_freq
            lo_wl, hi_wl = orig / lo, orig / hi
            scaled = torch.where(wavelen > lo_wl, inv_freq / factor, inv_freq)
            smooth = (orig / wavelen - lo) / (hi - lo)
            smoothed = (1 - smooth) / factor * inv_freq + smooth * inv_freq
            mid = (wavelen <= lo_wl) & (wavelen >= hi_wl)
            inv_freq = torch.where(mid, smoothed, scaled)
    t = torch.arange(max_len, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)  # [max_len, head_dim/2]
    emb = torch.cat([freqs, freqs], dim=-1)
    cos, sin = emb.cos().to(dtype), emb.sin().to(dtype)
    if device is not None:
        cos, sin = cos.to(device), sin.to(device)
    return cos, sin


def _rotate_half(x: torch.Tensor) -> torch.Tensor:
    half = x.shape[-1] // 2
    return torch.cat([-x[..., half:], x[..., :half]], dim=-1)


def apply_rope(
    q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor, position_ids: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    """q,k: [batch, heads, len, head_dim]; position_ids: [batch, len] or [len]."""
    if position_ids.dim() == 1:
        position_ids = position_ids.unsqueeze(0)
    cos_g = cos[position_ids].unsqueeze(1).to(q.dtype)  # [b, 1, len, hd]
    sin_g = sin[position_ids].unsqueeze(1).to(q.dtype)
    q_out = q * cos_g + _rotate_half(q) * sin_g
    k_out = k * cos_g + _rotate_half(k) * sin_g
    return q_out, k_out


def repeat_kv(x: torch.Tensor, n_rep: int) -> torch.Tensor:
    """[b, kv_heads, len, hd] -> [b, kv_heads*n_rep, len, hd]."""
    if n_rep == 1:
        return x
    b, kvh, slen, hd = x.shape
    return x[:, :, None].expand(b, kvh, n_rep, slen, hd).reshape(b, kvh * n_rep, slen, hd)


def attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    *,
    causal: bool,
    kv_offset: int = 0,
    attn_bias: Optional[torch.Tensor] = None,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Reference attention with fp32 softmax.

    q: [b, n_heads, q_len, hd]; k/v: [b, n_kv_heads, kv_len, hd].
    `causal` masks position (i + kv_offset) < j for query i, key j.
    `kv_offset` is the absolute position of q[0] within the kv sequence
    (kv_len = kv_offset + q_len when attending over a cache).
    `attn_bias`: optional additive bias [b or 1, n_heads, q_len, kv_len] (ALiBi).
    """
    b, n_heads, q_len, hd = q.shape
    n_kv = k.shape[1]
    if n_kv != n_heads:
        k = repeat_kv(k, n_heads // n_kv)
        v = repeat_kv(v, n_heads // n_kv)
    if scale is None:
        scale = 1.0 / math.sqrt(hd)
    scores = torch.matmul(q.to(torch.float32), k.to(torch.float32).transpose(-1, -2)) * scale
    if attn_bias is not None:
        scores = scores + attn_bias.to(torch.float32)
    if causal and q_len > 1:
        kv_len = k.shape[2]
        q_pos = torch.arange(q_len, device=q.device)[:, None] + kv_offset
        k_pos = torch.arange(kv_len, device=q.device)[None, :]
        mask = k_pos > q_pos
        scores = scores.masked_fill(mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    out = torch.matmul(probs.to(v.dtype), v)
    return out


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return F.silu(gate) * up


def gelu(x: torch.Tensor) -> torch.Tensor:
    return F.gelu(x, approximate="tanh")


def build_alibi_slopes(n_heads: int) -> torch.Tensor:
    """ALiBi per-head slopes (same closed form as the ALiBi paper)."""

    def pow2_slopes(n):
        start = 2.0 ** (-(2.0 ** -(math.log2(n) - 3)))
        return [start * (start**i) for i in range(n)]

    if math.log2(n_heads).is_integer():
        slopes = pow2_slopes(n_heads)
    else:
        closest = 2 ** math.floor(math.log2(n_heads))
        slopes = pow2_slopes(closest)
        extra = pow2_slopes(2 * closest)
        slopes += extra[0::2][: n_heads - closest]
    return torch.tensor(slopes, dtype=torch.float32)


def build_alibi_bias(
    n_heads: int, q_len: int, kv_len: int, device=None, dtype=torch.float32
) -> torch.Tensor:
    """[1, n_heads, q_len, kv_len] additive bias: slope * (j - i_abs) for j <= i."""
    slopes = build_alibi_slopes(n_heads).to(device)
    k_pos = torch.arange(kv_len, device=device, dtype=torch.float32)
    # bloom-style: bias depends only on key position distance from the last axis
    bias = slopes[:, None, None] * k_pos[None, None, :]  # [heads, 1, kv_len]
    return bias.unsqueeze(0).to(dtype).expand(1, n_heads, q_len, kv_len)
