"""Autoregressive generation against remote sessions.

Two paths, matching the reference's RemoteGenerationMixin capability
(reference client/remote_generation.py:20-163, which delegates to HF
GenerationMixin):

* the NATIVE loop (greedy / temperature / top-k / top-p / repetition penalty /
  simple beam search with server-side KV reorder, session resume) — the fast
  serving path, no per-token transformers overhead;
* full `transformers.GenerationMixin` delegation via `_HFGenerateFacade`
  whenever HF-specific machinery is requested (logits_processor,
  stopping_criteria, streamer, generation_config, constrained decoding, ...):
  a `RemotePastKeyValues` Cache shim tracks the server-side session position
  and beam reorders ship to the servers as hypo_ids.
"""

from __future__ import annotations

import contextlib
from typing import Optional

import torch
import torch.nn.functional as F

from petals_amd.client.inference_session import InferenceSession

# kwargs that route generate() through transformers.GenerationMixin
_HF_ONLY_KWARGS = frozenset(
    [
        "logits_processor", "stopping_criteria", "streamer", "generation_config",
        "prefix_allowed_tokens_fn", "synced_gpus", "assistant_model",
        "negative_prompt_ids", "negative_prompt_attention_mask",
        "bad_words_ids", "force_words_ids", "constraints", "num_beam_groups",
        "diversity_penalty", "penalty_alpha", "top_a", "typical_p", "epsilon_cutoff",
        "eta_cutoff", "no_repeat_ngram_size", "encoder_no_repeat_ngram_size",
        "min_length", "min_new_tokens", "exponential_decay_length_penalty",
        "suppress_tokens", "begin_suppress_tokens", "forced_bos_token_id",
        "forced_eos_token_id", "guidance_scale", "low_memory", "length_penalty",
        "early_stopping", "num_return_sequences", "output_scores", "output_logits",
        "return_dict_in_generate", "use_hf_generate",
    ]
)


class RemotePastKeyValues:
    """Pretends to be a transformers Cache: only tracks the number of tokens
    already consumed by the remote session (the real KV lives server-side).
    Parity: reference client/remote_generation.py:20-42."""

    is_compileable = False

    def __init__(self, seen: int = 0):
        self.layers = []  # transformers-5 Cache protocol
        self._seen = seen
        self.hypo_ids: Optional[torch.Tensor] = None

    def get_seq_length(self, layer_idx: int = 0) -> int:
        return self._seen

    def get_max_cache_shape(self):
        return None

    def get_max_length(self):
        return None

    def update_seen(self, n: int) -> None:
        self._seen += n

    def __len__(self):
        return 0

    def __getitem__(self, _):
        from petals_amd.utils.misc import DUMMY

        return [DUMMY]


_FACADE_CLS = None


def _get_facade_cls():
    """Lazily define the facade as a real transformers.GenerationMixin
    SUBCLASS (generate() looks decoding methods up on type(self))."""
    global _FACADE_CLS
    if _FACADE_CLS is not None:
        return _FACADE_CLS
    import transformers
    from transformers.modeling_outputs import CausalLMOutputWithPast

    class _HFGenerateFacade(transformers.GenerationMixin):
        """The object transformers.GenerationMixin.generate() runs against:
        it exposes the HF model surface (config/generation_config/forward/
        prepare_inputs_for_generation/_reorder_cache) over a Distributed*
        model while keeping the real model's own config untouched."""

        main_input_name = "input_ids"

        def __init__(self, model):
            self._model = model
            cfg = model.config
            self.config = transformers.PretrainedConfig(
                vocab_size=cfg.vocab_size,
                hidden_size=cfg.hidden_size,
                is_decoder=True,
                is_encoder_decoder=False,
            )
            self.generation_config = transformers.GenerationConfig(
                eos_token_id=getattr(cfg, "eos_token_id", None),
                bos_token_id=getattr(cfg, "bos_token_id", None),
                pad_token_id=getattr(cfg, "pad_token_id", None),
            )

        @property
        def device(self):
            return self._model.lm_head.weight.device

        def can_generate(self):
            return True

        def get_experts_implementation(self):
            return {}

        def set_experts_implementation(self, impl):
            pass

        def __call__(self, input_ids=None, past_key_values=None, **kwargs):
            if past_key_values is not None and past_key_values.hypo_ids is not None:
                self._model._next_hypo_ids = past_key_values.hypo_ids
                past_key_values.hypo_ids = None
            out = self._model(input_ids=input_ids)
            if past_key_values is not None:
                past_key_values.update_seen(input_ids.shape[1])
            return CausalLMOutputWithPast(logits=out.logits, past_key_values=past_key_values)

        forward = __call__

        def prepare_inputs_for_generation(self, input_ids, past_key_values=None, **kwargs):
            if past_key_values is not None and past_key_values.get_seq_length() > 0:
                input_ids = input_ids[:, past_key_values.get_seq_length() :]
            return {"input_ids": input_ids, "past_key_values": past_key_values}

        def _reorder_cache(self, past_key_values, beam_idx):
            # server-side KV reorder: ship beam_idx as hypo_ids with the next step
            past_key_values.hypo_ids = beam_idx.to(torch.int64).cpu()
            return past_key_values

    _FACADE_CLS = _HFGenerateFacade
    return _FACADE_CLS


class RemoteGenerationMixin:
    """Mixin for Distributed*ForCausalLM models."""

    @torch.inference_mode()
    def generate_hf(
        self,
        input_ids: Optional[torch.Tensor] = None,
        *,
        session: Optional[InferenceSession] = None,
        **kwargs,
    ) -> torch.Tensor:
        """Full `transformers.GenerationMixin.generate` against the swarm:
        logits processors, stopping criteria, streamers, beam variants and
        constrained decoding all work; the KV cache is the remote session.
        Parity: reference client/remote_generation.py:84-163."""
        kwargs.pop("use_hf_generate", None)
        assert input_ids is not None and input_ids.ndim == 2, "input_ids must be [batch, seq]"
        batch, prompt_len = input_ids.shape
        num_beams = kwargs.get("num_beams", 1)
        gc = kwargs.get("generation_config")
        if num_beams == 1 and gc is not None:
            num_beams = getattr(gc, "num_beams", 1) or 1
        max_new_tokens = kwargs.get("max_new_tokens")
        max_length = kwargs.get("max_length")
        if max_new_tokens is None and max_length is None and gc is not None:
            max_new_tokens = getattr(gc, "max_new_tokens", None)
            max_length = getattr(gc, "max_length", None)
        budget = (max_length - prompt_len) if (max_length is not None and max_new_tokens is None) else max_new_tokens
        assert budget is not None and budget > 0, "provide max_new_tokens or max_length"

        pre_seq_len = getattr(self, "pre_seq_len", 0) if getattr(self, "tuning_mode", None) else 0

        ctx = contextlib.nullcontext(session)
        if session is None:
            resumed = self.transformer.h.active_session
            if resumed is not None:
                want = batch * max(num_beams, 1)
                if getattr(resumed, "_batch_size", want) != want:
                    raise ValueError(
                        f"cannot resume the active inference session: it was opened with "
                        f"batch_size={resumed._batch_size} but generate() needs {want}"
                    )
                ctx = contextlib.nullcontext(resumed)
            else:
                ctx = self.transformer.h.inference_session(
                    max_length=pre_seq_len + prompt_len + budget,
                    batch_size=batch * max(num_beams, 1),
                )

        facade = _get_facade_cls()(self)
        with ctx as sess, self.transformer.h.use_session(sess):
            prev = sess.output_ids if sess.position > 0 else None
            if prev is not None:
                # resumed session: prepend the known context so HF criteria see
                # the full text; the past-cache shim skips it server-side
                input_ids = torch.cat([prev.to(input_ids.device), input_ids], dim=1)
            past = RemotePastKeyValues(seen=sess.position)
            out = facade.generate(input_ids, past_key_values=past, **kwargs)
            sess.output_ids = out if torch.is_tensor(out) else out.sequences
        return out

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
        if any(k in _HF_ONLY_KWARGS and kwargs[k] is not None for k in kwargs):
            # HF machinery requested: delegate to transformers.GenerationMixin
            hf_kwargs = dict(kwargs)
            for name, val in (
                ("max_new_tokens", max_new_tokens), ("max_length", max_length),
                ("do_sample", do_sample), ("temperature", temperature), ("top_k", top_k),
                ("top_p", top_p), ("repetition_penalty", repetition_penalty),
                ("num_beams", num_beams), ("eos_token_id", eos_token_id),
                ("pad_token_id", pad_token_id),
            ):
                if val is not None and not (name == "temperature" and val == 1.0) and not (
                    name in ("do_sample",) and val is False
                ) and not (name == "num_beams" and val == 1):
                    hf_kwargs[name] = val
            return self.generate_hf(input_ids, session=session, **hf_kwargs)
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
