"""TransformerBackend: wraps one decoder block for serving.

Parity with reference ``server/backend.py:24-235``: KV-cache descriptors,
chunked inference_step, beam-search cache reorder via hypo_ids, plus training
forward/backward used by rpc_forward / rpc_backward. Runs inside the
PriorityRuntime thread; handlers never touch the GPU directly.
"""

from __future__ import annotations

import logging
from typing import Dict, Optional, Sequence, Tuple

import torch

from petals_amd.data_structures import InferenceMetadata, ModuleUID
from petals_amd.server.memory_cache import MemoryCache, TensorDescriptor
from petals_amd.utils.misc import get_size_in_bytes, is_dummy

logger = logging.getLogger(__name__)


class TransformerBackend:
    def __init__(
        self,
        uid: ModuleUID,
        block: torch.nn.Module,
        *,
        config,
        memory_cache: MemoryCache,
        dtype: torch.dtype,
        max_chunk_size_bytes: int = 256 * 1024 * 1024,
    ):
        self.uid = uid
        self.block = block
        self.config = config
        self.memory_cache = memory_cache
        self.dtype = dtype
        self.dtype_bytes = get_size_in_bytes(dtype)
        self.max_chunk_size_bytes = max_chunk_size_bytes
        for p in block.parameters():
            p.requires_grad_(False)

    @property
    def device(self) -> torch.device:
        return next(self.block.parameters()).device

    def get_inference_cache_descriptors(self, batch_size: int, max_length: int) -> Sequence[TensorDescriptor]:
        k_shape, v_shape = self.block.kv_cache_shape(batch_size, max_length)
        return [TensorDescriptor(k_shape, self.dtype), TensorDescriptor(v_shape, self.dtype)]

    def cache_bytes_per_token(self, batch_size: int = 1) -> int:
        descs = self.get_inference_cache_descriptors(batch_size, 1)
        return sum(d.nbytes for d in descs)

    # ---------------------------------------------------------- inference

    @torch.inference_mode()
    def inference_step(
        self,
        hidden_states: torch.Tensor,
        hypo_ids: torch.Tensor,
        inference_info: InferenceMetadata,
    ) -> Tuple[torch.Tensor, ...]:
        assert hidden_states.ndim == 3
        seq_len = hidden_states.shape[1]
        prefix_length = inference_info.prefix_length
        with self.memory_cache.use_cache(*inference_info.cache_handles) as cache_tensors:
            k_cache, v_cache = cache_tensors
            if hypo_ids is not None and not is_dummy(hypo_ids):
                hypo_ids = hypo_ids.to(k_cache.device)
                k_cache[...] = k_cache[hypo_ids]
                v_cache[...] = v_cache[hypo_ids]
            max_chunk = self._estimate_max_chunk_length(hidden_states, prefix_length)
            if seq_len <= max_chunk:
                out = self.block(hidden_states, kv_cache=(k_cache, v_cache), prefix_length=prefix_length)
            else:
                out = torch.empty_like(hidden_states)
                for offset in range(0, seq_len, max_chunk):
                    chunk = hidden_states[:, offset : offset + max_chunk]
                    out[:, offset : offset + chunk.shape[1]] = self.block(
                        chunk, kv_cache=(k_cache, v_cache), prefix_length=prefix_length + offset
                    )
            return (out,)

    def _estimate_max_chunk_length(self, hidden_states: torch.Tensor, prefix_length: int) -> int:
        """Bound prefill chunks so attention score matrices fit reserved memory
        (parity: backend.py:146-152)."""
        batch_size, seq_length, _ = hidden_states.shape
        worst_case_length = prefix_length + seq_length
        n_heads = getattr(self.config, "num_attention_heads", 32)
        attn_bytes_per_token = n_heads * batch_size * 4 * worst_case_length  # fp32 scores
        return max(1, self.max_chunk_size_bytes // attn_bytes_per_token)

    # ----------------------------------------------------------- training

    @torch.inference_mode()
    def forward(self, hidden_states: torch.Tensor) -> torch.Tensor:
        """Stateless full-sequence causal forward (training fwd pass)."""
        return self.block(hidden_states)

    def backward(
        self,
        inputs: torch.Tensor,
        grad_outputs: torch.Tensor,
        prompt: Optional[torch.Tensor] = None,
    ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
        """Re-runs forward with grad enabled, returns (grad_inputs, grad_prompt).

        `prompt` (deep ptune) was ADDED to the first positions of `inputs`
        *before* this block on the forward pass; its gradient is the slice of
        grad_inputs over those positions (parity: block_functions.py:84-141)."""
        with torch.enable_grad():
            inputs = inputs.detach().clone().requires_grad_(True)
            if prompt is not None and not is_dummy(prompt):
                pre = prompt.shape[1]
                hidden = inputs.clone()
                hidden[:, :pre] += prompt
            else:
                hidden = inputs
            outputs = self.block(hidden)
            torch.autograd.backward([outputs], [grad_outputs])
        grad_inputs = inputs.grad
        grad_prompt = None
        if prompt is not None and not is_dummy(prompt):
            grad_prompt = grad_inputs[:, : prompt.shape[1]].clone()
        return grad_inputs, grad_prompt

    def get_info(self) -> Dict:
        return {
            "uid": self.uid,
            "hidden_size": self.config.hidden_size,
            "dtype": str(self.dtype).replace("torch.", ""),
            "device": str(self.device),
        }

    def shutdown(self):
        dummy = torch.tensor([])
        for p in self.block.parameters():
            p.data = dummy
