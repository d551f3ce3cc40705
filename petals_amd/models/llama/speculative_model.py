"""Speculative decoding over the swarm (parity: reference
models/llama/speculative_model.py:13-111).

A local draft model proposes `speculative_tokens` greedy continuations; the
remote model verifies them in ONE multi-token session step; accepted tokens
advance the session, rejected ones are rolled back server-side via the
session.position setter (start_from_position metadata -> KV-cache pointer
rewind, server/handler.py rpc_inference).

With greedy sampling the output is IDENTICAL to plain generate() regardless of
draft quality — only latency changes.
"""

from __future__ import annotations

from typing import Optional

import torch


class DistributedLlamaForSpeculativeGeneration:
    def __init__(self, model, draft_model):
        """model: DistributedLlamaForCausalLM; draft_model: any local callable
        `draft_model(input_ids).logits` (e.g. a small HF model)."""
        self.model = model
        self.draft_model = draft_model
        self.config = model.config

    @torch.inference_mode()
    def generate(
        self,
        input_ids: torch.Tensor,
        *,
        max_new_tokens: int,
        speculative_tokens: int = 4,
        eos_token_id: Optional[int] = None,
    ) -> torch.Tensor:
        assert input_ids.shape[0] == 1, "speculative decoding supports batch_size=1"
        seq = self.model.transformer.h
        prompt_len = input_ids.shape[1]
        max_length = prompt_len + max_new_tokens + speculative_tokens + 1

        generated = input_ids
        with seq.inference_session(max_length=max_length) as session, seq.use_session(session):
            # prefill -> first confirmed token
            logits = self.model(input_ids=input_ids).logits
            cur = logits[:, -1, :].argmax(dim=-1, keepdim=True)
            generated = torch.cat([generated, cur], dim=1)

            while generated.shape[1] - prompt_len < max_new_tokens:
                if eos_token_id is not None and generated[0, -1].item() == eos_token_id:
                    break
                k = min(speculative_tokens, prompt_len + max_new_tokens - generated.shape[1] + 1)
                # draft proposes k tokens greedily (local full-context recompute)
                draft_seq = generated
                for _ in range(k):
                    d_logits = self.draft_model(input_ids=draft_seq).logits
                    nxt = d_logits[:, -1, :].argmax(dim=-1, keepdim=True)
                    draft_seq = torch.cat([draft_seq, nxt], dim=1)
                draft_tokens = draft_seq[:, generated.shape[1]:]  # [1, k]

                # verify in one multi-token step: feed [cur, d1..d_{k-1}]
                pos_before = session.position
                verify_inputs = torch.cat([generated[:, -1:], draft_tokens[:, : k - 1]], dim=1)
                v_logits = self.model(input_ids=verify_inputs).logits  # [1, k, vocab]
                preds = v_logits.argmax(dim=-1)  # predicted token AFTER each input position

                n_acc = 0
                while n_acc < k - 1 and preds[0, n_acc].item() == draft_tokens[0, n_acc].item():
                    n_acc += 1
                next_token = preds[:, n_acc : n_acc + 1]

                accepted = draft_tokens[:, :n_acc]
                generated = torch.cat([generated, accepted, next_token], dim=1)
                # roll the remote KV caches back past the rejected suffix
                session.position = pos_before + n_acc + 1

        if generated.shape[1] > prompt_len + max_new_tokens:
            generated = generated[:, : prompt_len + max_new_tokens]
        return generated
