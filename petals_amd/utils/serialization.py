"""Tensor (de)serialization for the wire.

Parity with the reference's hivemind ``serialize_torch_tensor`` usage
(`client/remote_forward_backward.py:67`, `server/handler.py:128` in
/root/reference): a tensor travels as a small msgpack-able descriptor dict plus
a raw byte buffer, with optional lossy wire compression:

  * NONE      — raw bytes of the original dtype
  * FLOAT16   — cast fp32 -> fp16 on the wire (restored to the original dtype)
  * BFLOAT16  — cast fp32 -> bf16 on the wire
  * BLOCKWISE_8BIT — per-row absmax int8 (training-grade gradient compression)

All functions are CPU-side; callers move tensors to CPU before serializing.
"""

from __future__ import annotations

import struct
from typing import Any, Dict, List, Optional, Tuple

import torch

NONE = "none"
FLOAT16 = "float16"
BFLOAT16 = "bfloat16"
BLOCKWISE_8BIT = "blockwise_8bit"

_DTYPE_TO_STR = {
    torch.float32: "f32",
    torch.float16: "f16",
    torch.bfloat16: "bf16",
    torch.int8: "i8",
    torch.uint8: "u8",
    torch.int32: "i32",
    torch.int64: "i64",
    torch.bool: "bool",
}
_STR_TO_DTYPE = {v: k for k, v in _DTYPE_TO_STR.items()}


def _raw_bytes(t: torch.Tensor) -> bytes:
    t = t.detach().contiguous().cpu()
    if t.dtype == torch.bfloat16:
        t = t.view(torch.int16)
    if t.dtype == torch.bool:
        t = t.to(torch.uint8)
    return t.numpy().tobytes()


def _from_raw(buf: bytes, dtype: torch.dtype, shape: Tuple[int, ...]) -> torch.Tensor:
    import numpy as np

    if dtype == torch.bfloat16:
        arr = np.frombuffer(bytearray(buf), dtype=np.int16)
        t = torch.from_numpy(arr).view(torch.bfloat16)
    elif dtype == torch.bool:
        arr = np.frombuffer(bytearray(buf), dtype=np.uint8)
        t = torch.from_numpy(arr).to(torch.bool)
    else:
        np_dtype = torch.empty(0, dtype=dtype).numpy().dtype
        arr = np.frombuffer(bytearray(buf), dtype=np_dtype)
        t = torch.from_numpy(arr)
    return t.reshape(shape)


def serialize_tensor(
    tensor: torch.Tensor, compression: str = NONE
) -> Tuple[Dict[str, Any], bytes]:
    """Returns (descriptor, payload). The descriptor is msgpack-friendly."""
    requires_grad = bool(tensor.requires_grad)
    tensor = tensor.detach()
    orig_dtype = _DTYPE_TO_STR[tensor.dtype]
    desc: Dict[str, Any] = {
        "shape": list(tensor.shape),
        "dtype": orig_dtype,
        "compression": compression,
        "requires_grad": requires_grad,
    }
    if compression == NONE:
        return desc, _raw_bytes(tensor)
    if compression == FLOAT16:
        return desc, _raw_bytes(tensor.to(torch.float16))
    if compression == BFLOAT16:
        return desc, _raw_bytes(tensor.to(torch.bfloat16))
    if compression == BLOCKWISE_8BIT:
        flat = tensor.to(torch.float32).reshape(-1)
        n = flat.numel()
        block = 4096
        pad = (-n) % block
        if pad:
            flat = torch.cat([flat, flat.new_zeros(pad)])
        rows = flat.reshape(-1, block)
        absmax = rows.abs().amax(dim=1).clamp_min(1e-12)
        q = torch.clamp((rows / absmax[:, None]) * 127.0, -127, 127).round().to(torch.int8)
        payload = struct.pack("<q", n) + _raw_bytes(absmax) + _raw_bytes(q)
        return desc, payload
    raise ValueError(f"unknown compression {compression!r}")


def deserialize_tensor(desc: Dict[str, Any], payload: bytes) -> torch.Tensor:
    shape = tuple(desc["shape"])
    orig_dtype = _STR_TO_DTYPE[desc["dtype"]]
    compression = desc.get("compression", NONE)
    if compression == NONE:
        t = _from_raw(payload, orig_dtype, shape)
    elif compression == FLOAT16:
        t = _from_raw(payload, torch.float16, shape).to(orig_dtype)
    elif compression == BFLOAT16:
        t = _from_raw(payload, torch.bfloat16, shape).to(orig_dtype)
    elif compression == BLOCKWISE_8BIT:
        (n,) = struct.unpack("<q", payload[:8])
        block = 4096
        nrows = (n + block - 1) // block
        absmax_bytes = nrows * 4
        absmax = _from_raw(payload[8 : 8 + absmax_bytes], torch.float32, (nrows,))
        q = _from_raw(payload[8 + absmax_bytes :], torch.int8, (nrows, block))
        flat = (q.to(torch.float32) / 127.0) * absmax[:, None]
        t = flat.reshape(-1)[:n].reshape(shape).to(orig_dtype)
    else:
        raise ValueError(f"unknown compression {compression!r}")
    if desc.get("requires_grad"):
        t.requires_grad_(True)
    return t


def serialize_tensors(
    tensors: List[torch.Tensor], compressions: Optional[List[str]] = None
) -> Tuple[List[Dict[str, Any]], List[bytes]]:
    if compressions is None:
        compressions = [NONE] * len(tensors)
    descs, bufs = [], []
    for t, c in zip(tensors, compressions):
        d, b = serialize_tensor(t, c)
        descs.append(d)
        bufs.append(b)
    return descs, bufs


def deserialize_tensors(descs: List[Dict[str, Any]], bufs: List[bytes]) -> List[torch.Tensor]:
    return [deserialize_tensor(d, b) for d, b in zip(descs, bufs)]
