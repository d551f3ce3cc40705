"""Autoregressive generation against remote sessions.

Capability parity with the reference's RemoteGenerationMixin
(client/remote_generation.py:84 — which delegates to HF GenerationMixin):
greedy, temperature/top-k/top-p sampling, repetition penalty, simple beam
search (server-side KV reorder via hypo_ids), resuming an open session across
multiple generate() calls. Implemented natively to avoid coupling the client
to transformers' generation internals.
"""

from __future__ import annotations

import contextlib
from typing import Optional

import torch
import torch.nn.functional as F

from petals_amd.client.inference_session import InferenceSession


class RemoteGenerationMixin:
    """Mixin for Distributed*ForCausalLM models."""

    @torch.inference_mode()
    def generate(
        self,
        input_ids: Optional[torch.Tensor] = None,
        *,
        max_new_tokens: Optional[int] = None,
        max_length: Optional[int] = None,
        do_sample: bool = False,
        temperature: float = 1.0,
        top_k: Optional[int] = None,
        top_p: Optional[float] = None,
        repetition_penalty: Optional[float] = None,
        num_beams: int = 1,
        eos_token_id: Optional[int] = None,
        pad_token_id: Optional[int] = None,
        session: Optional[InferenceSession] = None,
        **kwargs,
    ) -> torch.Tensor:
        assert input_ids is not None and input_ids.ndim == 2, "input_ids must be [batch, seq]"
        if eos_token_id is None:
            eos_token_id = getattr(self.config, "eos_token_id", None)
        batch, prompt_len = input_ids.shape

        if max_new_tokens is None:
            assert max_length is not None, "provide max_new_tokens or max_length"
            max_new_tokens = max_length - prompt_len
        assert max_new_tokens > 0

        if num_beams > 1:
            return self._beam_search(
                input_ids, max_new_tokens=max_new_tokens, num_beams=num_beams, eos_token_id=eos_token_id
            )

        # ptune prefix counts against session length
        pre_seq_len = getattr(self, "pre_seq_len", 0) if getattr(self, "tuning_mode", None) else 0

        ctx = contextlib.nullcontext(session)
        if session is None:
            resumed = self.transformer.h.active_session
            if resumed is not None:
                ctx = contextlib.nullcontext(resumed)
            else:
                ctx = self.transformer.h.inference_session(
                    max_length=pre_seq_len + prompt_len + max_new_tokens, batch_size=batch
                )

        with ctx as sess, self.transformer.h.use_session(sess):
            generated = input_ids
            unfinished = torch.ones(batch, dtype=torch.bool, device=input_ids.device)
            step_input = input_ids
            for _ in range(max_new_tokens):
                logits = self(input_ids=step_input).logits[:, -1, :]
                if repetition_penalty is not None and repetition_penalty != 1.0:
                    logits = self._apply_repetition_penalty(logits, generated, repetition_penalty)
                if do_sample:
                    next_token = self._sample(logits, temperature, top_k, top_p)
                else:
                    next_token = logits.argmax(dim=-1)
                next_token = next_token.to(generated.device)
                if eos_token_id is not None:
                    if pad_token_id is not None:
                        next_token = torch.where(
                            unfinished, next_token, torch.full_like(next_token, pad_token_id)
                        )
                    unfinished &= next_token != eos_token_id
                generated = torch.cat([generated, next_token[:, None]], dim=1)
                step_input = next_token[:, None]
                if eos_token_id is not None and not unfinished.any():
                    break
            sess.output_ids = generated
        return generated

    @staticmethod
    def _sample(logits: torch.Tensor, temperature: float, top_k: Optional[int], top_p: Optional[float]):
        if temperature != 1.0:
            logits = logits / max(temperature, 1e-6)
        if top_k is not None and top_k > 0:
            kth = torch.topk(logits, min(top_k, logits.shape[-1]), dim=-1).values[:, -1:]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        if top_p is not None and 0 < top_p < 1.0:
            sorted_logits, sorted_idx = torch.sort(logits, descending=True, dim=-1)
            cum = torch.softmax(sorted_logits, dim=-1).cumsum(dim=-1)
            mask = cum - torch.softmax(sorted_logits, dim=-1) > top_p
            sorted_logits = sorted_logits.masked_fill(mask, float("-inf"))
            logits = torch.full_like(logits, float("-inf")).scatter(1, sorted_idx, sorted_logits)
        probs = torch.softmax(logits.float(), dim=-1)
        return torch.multinomial(probs, 1).squeeze(-1)

    @staticmethod
    def _apply_repetition_penalty(logits: torch.Tensor, generated: torch.Tensor, penalty: float):
        score = torch.gather(logits, 1, generated)
        score = torch.where(score < 0, score * penalty, score / penalty)
        return logits.scatter(1, generated, score)

    @torch.inference_mode()
    def _beam_search(self, input_ids: torch.Tensor, *, max_new_tokens: int, num_beams: int, eos_token_id):
        """Basic beam search; beams live server-side as batch rows, reordered
        in the KV caches via hypo_ids each step."""
        batch, prompt_len = input_ids.shape
        assert batch == 1, "beam search currently supports batch_size=1"
        pre_seq_len = getattr(self, "pre_seq_len", 0) if getattr(self, "tuning_mode", None) else 0
        expanded = input_ids.expand(num_beams, prompt_len).contiguous()

        with self.transformer.h.inference_session(
            max_length=pre_seq_len + prompt_len + max_new_tokens, batch_size=num_beams
        ) as sess, self.transformer.h.use_session(sess):
            logits = self(input_ids=expanded).logits[:, -1, :].float()
            logprobs = F.log_softmax(logits[0:1], dim=-1)  # all beams identical on step 1
            scores, next_tokens = logprobs.topk(num_beams, dim=-1)
            beam_scores = scores[0]
            sequences = torch.cat([expanded, next_tokens[0][:, None]], dim=1)
            step_input = next_tokens[0][:, None]

            for _ in range(max_new_tokens - 1):
                logits = self(input_ids=step_input).logits[:, -1, :].float()
                logprobs = F.log_softmax(logits, dim=-1)
                total = beam_scores[:, None] + logprobs  # [beams, vocab]
                vocab = total.shape[-1]
                flat = total.reshape(-1)
                beam_scores, flat_idx = flat.topk(num_beams)
                beam_idx = flat_idx // vocab
                token_idx = flat_idx % vocab
                sequences = torch.cat([sequences[beam_idx], token_idx[:, None]], dim=1)
                step_input = token_idx[:, None]
                # reorder server-side caches to match the chosen beams
                self._next_hypo_ids = beam_idx.to(torch.int64)
                if eos_token_id is not None and (token_idx == eos_token_id).all():
                    break
            best = beam_scores.argmax()
            return sequences[best : best + 1]
