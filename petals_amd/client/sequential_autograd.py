"""Fault-tolerant autograd over a chain of remote servers.

Parity: reference client/sequential_autograd.py — forward splits the batch
into <=1024-token micro-batches processed concurrently (pipelined fill);
per-span activations are kept client-side so backward can resume after a
server failure by re-running forward only for the lost span.
"""

from __future__ import annotations

import asyncio
import logging
from typing import Any, List, Optional, Sequence, Tuple

import torch

from petals_amd.client.remote_forward_backward import run_remote_backward, run_remote_forward
from petals_amd.client.routing.sequence_manager import RemoteSequenceManager
from petals_amd.data_structures import RemoteSpanInfo
from petals_amd.utils.misc import DUMMY, is_dummy

logger = logging.getLogger(__name__)

MAX_TOKENS_IN_BATCH = 1024


async def sequential_forward(
    sequence_manager: RemoteSequenceManager,
    inputs: torch.Tensor,
    prompts: torch.Tensor,
    start_index: int = 0,
    end_index: Optional[int] = None,
) -> Tuple[torch.Tensor, List[torch.Tensor], List[RemoteSpanInfo]]:
    """Returns (outputs, per-span input activations, chosen spans)."""
    assert inputs.ndim == 3
    end_index = end_index if end_index is not None else sequence_manager.num_blocks

    block_idx = start_index
    intermediate_inputs: List[torch.Tensor] = []
    done_spans: List[RemoteSpanInfo] = []
    outputs = inputs
    sequences: List[RemoteSpanInfo] = []

    attempt = 0
    while block_idx < end_index:
        if not sequences or attempt > 0:
            sequences = sequence_manager.make_sequence(block_idx, end_index, mode="max_throughput")
        span = sequences.pop(0)
        try:
            uids = sequence_manager.block_uids[span.start : span.end]
            span_prompts = prompts[span.start : span.end] if not is_dummy(prompts) else DUMMY
            metadata = sequence_manager.get_request_metadata("rpc_forward", uids)
            addr = sequence_manager.address_of(span.peer_id)
            out = await run_remote_forward(
                sequence_manager.p2p,
                addr,
                uids,
                outputs,
                span_prompts,
                metadata=metadata,
                timeout=sequence_manager.config.request_timeout,
                compression=sequence_manager.config.wire_compression,
            )
            assert out.shape == outputs.shape, f"bad output shape {out.shape} vs {outputs.shape}"
            intermediate_inputs.append(outputs)
            done_spans.append(span)
            outputs = out
            block_idx = span.end
            sequence_manager.on_request_success(span.peer_id)
            attempt = 0
        except Exception as e:  # noqa: BLE001
            attempt += 1
            max_retries = sequence_manager.config.max_retries
            if max_retries is not None and attempt > max_retries:
                raise
            delay = sequence_manager.get_retry_delay(attempt)
            logger.warning("forward via %s failed (%r); retrying in %.1f s", span.peer_id[:8], e, delay)
            sequence_manager.on_request_failure(span.peer_id)
            await asyncio.sleep(delay)
    return outputs, intermediate_inputs, done_spans


async def sequential_backward(
    sequence_manager: RemoteSequenceManager,
    grad_outputs: torch.Tensor,
    intermediate_inputs: List[torch.Tensor],
    prompts: torch.Tensor,
    forward_sequences: List[RemoteSpanInfo],
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (grad_inputs, grad_prompts [num_blocks,...] or DUMMY)."""
    assert len(intermediate_inputs) == len(forward_sequences)
    intermediate_inputs = list(intermediate_inputs)
    forward_sequences = list(forward_sequences)
    grad_prompts_reversed: List[torch.Tensor] = []

    attempt = 0
    while forward_sequences and intermediate_inputs:
        inputs = intermediate_inputs.pop()
        span = forward_sequences.pop()
        while True:
            try:
                uids = sequence_manager.block_uids[span.start : span.end]
                span_prompts = prompts[span.start : span.end] if not is_dummy(prompts) else DUMMY
                metadata = sequence_manager.get_request_metadata("rpc_backward", uids)
                addr = sequence_manager.address_of(span.peer_id)
                grad_outputs, grad_prompts = await run_remote_backward(
                    sequence_manager.p2p,
                    addr,
                    uids,
                    inputs,
                    grad_outputs,
                    span_prompts,
                    metadata=metadata,
                    timeout=sequence_manager.config.request_timeout,
                    compression=sequence_manager.config.wire_compression,
                )
                if not is_dummy(grad_prompts):
                    grad_prompts_reversed.append(grad_prompts)
                elif not is_dummy(prompts):
                    grad_prompts_reversed.append(torch.zeros_like(prompts[span.start : span.end]))
                sequence_manager.on_request_success(span.peer_id)
                attempt = 0
                break
            except Exception as e:  # noqa: BLE001
                attempt += 1
                max_retries = sequence_manager.config.max_retries
                if max_retries is not None and attempt > max_retries:
                    raise
                delay = sequence_manager.get_retry_delay(attempt)
                logger.warning("backward via %s failed (%r); retrying in %.1f s", span.peer_id[:8], e, delay)
                sequence_manager.on_request_failure(span.peer_id)
                await asyncio.sleep(delay)
                # the failed span's activations may be lost: recompute forward for it
                _, new_inters, new_spans = await sequential_forward(
                    sequence_manager, inputs, prompts, start_index=span.start, end_index=span.end
                )
                assert len(new_inters) > 0 and len(new_spans) > 0
                # substitute the failed span with the new sub-chain; grad_outputs
                # corresponds to the LAST sub-span's output, so backward must
                # proceed deepest-sub-span first: push all recomputed entries and
                # pop the last one
                intermediate_inputs.extend(new_inters)
                forward_sequences.extend(new_spans)
                inputs = intermediate_inputs.pop()
                span = forward_sequences.pop()

    grad_prompts = (
        torch.cat(list(reversed(grad_prompts_reversed)), dim=0) if grad_prompts_reversed else DUMMY
    )
    return grad_outputs, grad_prompts


async def _gather_forward(sequence_manager, input_batches, prompt_batches):
    return await asyncio.gather(
        *(sequential_forward(sequence_manager, inp, pr) for inp, pr in zip(input_batches, prompt_batches))
    )


async def _gather_backward(sequence_manager, grad_batches, inter_batches, prompt_batches, span_batches):
    return await asyncio.gather(
        *(
            sequential_backward(sequence_manager, g, inters, pr, spans)
            for g, inters, pr, spans in zip(grad_batches, inter_batches, prompt_batches, span_batches)
        )
    )


class _RemoteSequentialAutogradFunction(torch.autograd.Function):
    """Splits batches into <=1024-token micro-batches and routes them through
    the swarm concurrently (parity: reference :223-277)."""

    @staticmethod
    def forward(ctx, inputs: torch.Tensor, prompts: torch.Tensor, sequence_manager: RemoteSequenceManager):
        batch_size = max(MAX_TOKENS_IN_BATCH // max(inputs.shape[1], 1), 1)
        input_batches: Sequence[torch.Tensor] = inputs.detach().split(batch_size)
        if is_dummy(prompts):
            prompt_batches = [DUMMY] * len(input_batches)
        else:
            prompt_batches = prompts.detach().split(batch_size, dim=1)

        results = sequence_manager.run_coroutine(
            _gather_forward(sequence_manager, input_batches, prompt_batches)
        )
        assert len(results) == len(input_batches)

        output_batches = [r[0] for r in results]
        intermediate_input_batches = [r[1] for r in results]
        sequences_for_batches = [r[2] for r in results]

        ctx.prompt_batches = prompt_batches
        ctx.sequence_manager = sequence_manager
        ctx.intermediate_input_batches = intermediate_input_batches
        ctx.sequences_for_batches = sequences_for_batches
        return torch.cat(output_batches, dim=0)

    @staticmethod
    def backward(ctx, grad_outputs: torch.Tensor):
        intermediate_input_batches: List[Sequence[torch.Tensor]] = ctx.intermediate_input_batches
        forward_sequences: List[Sequence[Any]] = ctx.sequences_for_batches
        sequence_manager = ctx.sequence_manager

        # must mirror forward's split boundaries exactly (same formula), else
        # grad splits diverge from input splits when the batch doesn't divide evenly
        batch_size = max(MAX_TOKENS_IN_BATCH // max(grad_outputs.shape[1], 1), 1)
        grad_output_batches: Sequence[torch.Tensor] = grad_outputs.split(batch_size)
        assert len(grad_output_batches) == len(intermediate_input_batches)

        results = sequence_manager.run_coroutine(
            _gather_backward(
                sequence_manager, grad_output_batches, intermediate_input_batches, ctx.prompt_batches, forward_sequences
            )
        )
        grad_input_batches = [r[0] for r in results]
        grad_prompt_batches = [r[1] for r in results]

        grad_inputs = torch.cat(grad_input_batches, dim=0)
        dummy_grad_prompts = [is_dummy(g) for g in grad_prompt_batches]
        grad_prompts = torch.cat(grad_prompt_batches, dim=1) if not any(dummy_grad_prompts) else None
        return (grad_inputs, grad_prompts, None)
