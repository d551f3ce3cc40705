"""Client-local output projection (parity: client/lm_head.py — incl. the
chunked matmul fallback for low-memory CPUs)."""

from __future__ import annotations

import dataclasses

import torch
import torch.nn.functional as F
from torch import nn


@dataclasses.dataclass
class LMHeadConfig:
    use_chunked_forward: str | bool = "auto"
    chunked_forward_step: int = 16384


class LMHead(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.weight = nn.Parameter(torch.zeros(config.vocab_size, config.hidden_size), requires_grad=False)
        use_chunked = getattr(config, "use_chunked_forward", "auto")
        if use_chunked == "auto":
            # chunk on CPU fp32 to bound peak memory; never on GPU
            use_chunked = True
        self.use_chunked_forward = bool(use_chunked)
        self.chunked_forward_step = getattr(config, "chunked_forward_step", 16384)

    def forward(self, hidden_states: torch.Tensor) -> torch.Tensor:
        if (
            self.weight.dtype in (torch.float16, torch.bfloat16)
            and self.weight.device.type == "cpu"
            and self.use_chunked_forward
        ):
            return self.chunked_forward(hidden_states)
        return F.linear(hidden_states, self.weight.to(hidden_states.dtype))

    def chunked_forward(self, hidden_states: torch.Tensor) -> torch.Tensor:
        """fp32 matmul in vocab chunks: avoids materializing a full fp32 copy
        of a half-precision head on CPU."""
        hidden = hidden_states.float()
        out = torch.empty(*hidden.shape[:-1], self.weight.shape[0], dtype=torch.float32)
        step = self.chunked_forward_step
        for i in range(0, self.weight.shape[0], step):
            chunk = self.weight[i : i + step].float()
            out[..., i : i + step] = hidden @ chunk.T
        return out.to(hidden_states.dtype)
