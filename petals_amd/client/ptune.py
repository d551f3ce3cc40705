"""Prompt tuning: client-held trainable prompts (parity: client/ptune.py).

`tuning_mode="ptune"` prepends pre_seq_len trainable embeddings to the input;
`"deep_ptune"` additionally ships per-block prompts with every request (the
server ADDS them to the first pre_seq_len positions of that block's input).
"""

from __future__ import annotations

import dataclasses
from typing import Optional, Tuple

import torch
from torch import nn

from petals_amd.utils.misc import DUMMY


@dataclasses.dataclass
class PTuneConfig:
    pre_seq_len: int = 0
    tuning_mode: Optional[str] = None  # None | "ptune" | "deep_ptune"


class PTuneMixin:
    _keys_to_ignore_on_load_missing = [r"(intermediate_)?prompt_embeddings\.weight$"]

    def init_prompts(self, config) -> None:
        if config.tuning_mode and "ptune" in config.tuning_mode:
            assert config.pre_seq_len > 0, "pre_seq_len must be positive for ptune"
            self.pre_seq_len = config.pre_seq_len
            self.prefix_tokens = torch.arange(self.pre_seq_len).long()
            self.prompt_embeddings = nn.Embedding(self.pre_seq_len, config.hidden_size)
            if config.tuning_mode == "deep_ptune":
                self.intermediate_prompt_embeddings = nn.Embedding(
                    self.pre_seq_len, config.num_blocks * config.hidden_size
                )
                self.intermediate_prompt_embeddings.weight.data.zero_()
        elif config.tuning_mode:
            raise NotImplementedError(f"tuning_mode={config.tuning_mode} is not supported")
        self.tuning_mode = config.tuning_mode

    def get_prompt(self, batch_size: int) -> Tuple[torch.Tensor, torch.Tensor]:
        """Returns (input prompts [batch, pre_seq, hidden],
        intermediate prompts [num_blocks, batch, pre_seq, hidden] or DUMMY)."""
        prefix_tokens = self.prefix_tokens.unsqueeze(0).expand(batch_size, -1)
        prefix_tokens = prefix_tokens.to(self.prompt_embeddings.weight.device)
        prompts = self.prompt_embeddings(prefix_tokens)
        if self.tuning_mode == "deep_ptune":
            intermediate = self.intermediate_prompt_embeddings(prefix_tokens)
            intermediate = intermediate.view(
                batch_size, self.pre_seq_len, -1, prompts.shape[-1]
            ).permute(2, 0, 1, 3)  # [num_blocks, batch, pre_seq, hidden]
            return prompts, intermediate
        return prompts, DUMMY
