"""Chunked tensor streaming over an RpcStream.

Parity with the reference's hivemind streaming (`client/remote_forward_backward.py:41-64`,
`server/handler.py:128`): payloads larger than MAX_UNARY_PAYLOAD_SIZE travel as
a sequence of per-tensor byte chunks and are reassembled on the other side.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Tuple

import torch

from petals_amd.p2p.transport import RpcMessage, RpcStream
from petals_amd.utils import serialization

DEFAULT_CHUNK_BYTES = 16 << 20  # 16 MiB per frame part


async def send_tensors_streamed(
    stream: RpcStream,
    tensors: List[torch.Tensor],
    meta: Optional[Dict[str, Any]] = None,
    compressions: Optional[List[str]] = None,
    chunk_bytes: int = DEFAULT_CHUNK_BYTES,
    close: bool = True,
) -> None:
    """Sends `meta` + tensors as a chunk sequence, then (optionally) closes."""
    descs, bufs = serialization.serialize_tensors(tensors, compressions)
    header = RpcMessage(meta={**(meta or {}), "_stream_descs": descs, "_stream_n": len(bufs)})
    await stream.send(header)
    for i, buf in enumerate(bufs):
        for off in range(0, max(len(buf), 1), chunk_bytes):
            part = buf[off : off + chunk_bytes]
            last = off + chunk_bytes >= len(buf)
            await stream.send(
                RpcMessage(meta={"_chunk_of": i, "_last": last, "_bytes": part})
            )
    if close:
        await stream.close(RpcMessage(meta={"_stream_done": True}))


async def receive_tensors_streamed(stream: RpcStream, timeout: float = 120.0) -> Tuple[Dict[str, Any], List[torch.Tensor]]:
    """Receives one streamed tensor sequence (header + chunks)."""
    header = await stream.receive(timeout=timeout)
    meta = dict(header.meta)
    descs = meta.pop("_stream_descs")
    n = meta.pop("_stream_n")
    bufs: List[bytearray] = [bytearray() for _ in range(n)]
    done = [False] * n
    while not all(done):
        msg = await stream.receive(timeout=timeout)
        i = msg.meta.get("_chunk_of")
        if i is None:
            continue
        bufs[i].extend(msg.meta["_bytes"])
        if msg.meta.get("_last"):
            done[i] = True
    tensors = serialization.deserialize_tensors(descs, [bytes(b) for b in bufs])
    return meta, tensors
