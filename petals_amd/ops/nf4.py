"""CPU/torch reference for the NF4 format (numerics baseline for nf4.hip).

Format: see ops/csrc/nf4.hip — blocksize 64 along the last dim, bf16 absmax,
two nibbles per byte (even element = low nibble).
"""

from __future__ import annotations

from typing import Tuple

import torch

NF4_LEVELS = torch.tensor(
    [
        -1.0, -0.6961928009986877, -0.5250730514526367, -0.39491748809814453,
        -0.28444138169288635, -0.18477343022823334, -0.09105003625154495, 0.0,
        0.07958029955625534, 0.16093020141124725, 0.24611230194568634,
        0.33791524171829224, 0.4407098293304443, 0.5626170039176941,
        0.7229568362236023, 1.0,
    ],
    dtype=torch.float32,
)


def quantize(w: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """w: [in, out] (out % 64 == 0) -> (packed u8 [in, out/2], absmax bf16 [in, out/64])."""
    assert w.dim() == 2 and w.shape[1] % 64 == 0
    in_dim, out_dim = w.shape
    blocks = w.to(torch.float32).reshape(in_dim, out_dim // 64, 64)
    absmax = blocks.abs().amax(dim=-1).to(torch.bfloat16)
    scale = absmax.to(torch.float32).clamp_min(1e-12)
    normed = blocks / scale[..., None]
    # nearest level
    dists = (normed[..., None] - NF4_LEVELS.view(1, 1, 1, 16)).abs()
    idx = dists.argmin(dim=-1).to(torch.uint8)  # [in, out/64, 64]
    idx = idx.reshape(in_dim, out_dim)
    packed = (idx[:, 0::2] | (idx[:, 1::2] << 4)).contiguous()
    return packed, absmax.contiguous()


def dequantize(packed: torch.Tensor, absmax: torch.Tensor) -> torch.Tensor:
    in_dim, half = packed.shape
    out_dim = half * 2
    idx = torch.empty(in_dim, out_dim, dtype=torch.long)
    idx[:, 0::2] = (packed & 0xF).long()
    idx[:, 1::2] = (packed >> 4).long()
    vals = NF4_LEVELS[idx]
    scale = absmax.to(torch.float32).repeat_interleave(64, dim=1)
    return (vals * scale).to(torch.bfloat16)
