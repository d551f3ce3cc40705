"""RPC client for training forward/backward calls to one server span.

Parity: reference client/remote_forward_backward.py (unary vs stream choice at
MAX_UNARY_PAYLOAD_SIZE/2, :107).
"""

from __future__ import annotations

from typing import Dict, Optional, Sequence, Tuple

import torch

from petals_amd.data_structures import CHAIN_DELIMITER, ModuleUID
from petals_amd.p2p.streaming import receive_tensors_streamed, send_tensors_streamed
from petals_amd.p2p.transport import MAX_UNARY_PAYLOAD_SIZE, P2PNode, RpcMessage
from petals_amd.utils.misc import get_size_in_bytes


def _payload_bytes(tensors: Sequence[torch.Tensor]) -> int:
    return sum(t.numel() * get_size_in_bytes(t.dtype) for t in tensors)


async def run_remote_forward(
    p2p: P2PNode,
    addr: Tuple[str, int],
    uids: Sequence[ModuleUID],
    hidden_states: torch.Tensor,
    prompts: torch.Tensor,
    *,
    metadata: Optional[Dict] = None,
    timeout: float = 180.0,
    compression: str = "none",
) -> torch.Tensor:
    tensors = [hidden_states.cpu(), prompts.cpu()]
    meta = {"uids": CHAIN_DELIMITER.join(uids), **(metadata or {})}
    compressions = [compression] * len(tensors) if compression and compression != "none" else None
    if _payload_bytes(tensors) > MAX_UNARY_PAYLOAD_SIZE // 2:
        stream = await p2p.open_stream(addr, "petals.rpc_forward_stream", RpcMessage(meta=meta), timeout=timeout)
        await send_tensors_streamed(stream, tensors, compressions=compressions, close=True)
        _, outs = await receive_tensors_streamed(stream, timeout=timeout)
    else:
        resp = await p2p.call_unary(
            addr, "petals.rpc_forward", RpcMessage(meta=meta, tensors=tensors), timeout=timeout,
            compressions=compressions,
        )
        outs = resp.tensors
    return outs[0]


async def run_remote_backward(
    p2p: P2PNode,
    addr: Tuple[str, int],
    uids: Sequence[ModuleUID],
    inputs: torch.Tensor,
    grad_outputs: torch.Tensor,
    prompts: torch.Tensor,
    *,
    metadata: Optional[Dict] = None,
    timeout: float = 180.0,
    compression: str = "none",
) -> Tuple[torch.Tensor, torch.Tensor]:
    tensors = [inputs.cpu(), grad_outputs.cpu(), prompts.cpu()]
    meta = {"uids": CHAIN_DELIMITER.join(uids), **(metadata or {})}
    compressions = [compression] * len(tensors) if compression and compression != "none" else None
    if _payload_bytes(tensors) > MAX_UNARY_PAYLOAD_SIZE // 2:
        stream = await p2p.open_stream(addr, "petals.rpc_backward_stream", RpcMessage(meta=meta), timeout=timeout)
        await send_tensors_streamed(stream, tensors, compressions=compressions, close=True)
        _, outs = await receive_tensors_streamed(stream, timeout=timeout)
    else:
        resp = await p2p.call_unary(
            addr, "petals.rpc_backward", RpcMessage(meta=meta, tensors=tensors), timeout=timeout,
            compressions=compressions,
        )
        outs = resp.tensors
    return outs[0], outs[1]
