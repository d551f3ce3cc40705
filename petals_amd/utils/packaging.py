"""Flatten nested (args, kwargs) containing tensors into a tensor list + a
msgpack-able structure with ``"__T{i}"`` placeholders.

Parity: reference ``utils/packaging.py:21-49`` (pack_args_kwargs /
unpack_args_kwargs) — repeated tensors are deduplicated.
"""

from __future__ import annotations

from typing import Any, Dict, List, Tuple

import torch


def _pack(obj: Any, tensors: List[torch.Tensor], index: Dict[int, int]) -> Any:
    if isinstance(obj, torch.Tensor):
        key = id(obj)
        if key not in index:
            index[key] = len(tensors)
            tensors.append(obj)
        return f"__T{index[key]}"
    if isinstance(obj, (list, tuple)):
        packed = [_pack(x, tensors, index) for x in obj]
        return {"__tuple": packed} if isinstance(obj, tuple) else packed
    if isinstance(obj, dict):
        return {k: _pack(v, tensors, index) for k, v in obj.items()}
    return obj


def _unpack(obj: Any, tensors: List[torch.Tensor]) -> Any:
    if isinstance(obj, str) and obj.startswith("__T"):
        return tensors[int(obj[3:])]
    if isinstance(obj, dict):
        if set(obj.keys()) == {"__tuple"}:
            return tuple(_unpack(x, tensors) for x in obj["__tuple"])
        return {k: _unpack(v, tensors) for k, v in obj.items()}
    if isinstance(obj, list):
        return [_unpack(x, tensors) for x in obj]
    return obj


def pack_args_kwargs(*args, **kwargs) -> Tuple[List[torch.Tensor], Any]:
    """Returns (flat_tensors, structure)."""
    tensors: List[torch.Tensor] = []
    index: Dict[int, int] = {}
    structure = _pack((list(args), kwargs), tensors, index)
    return tensors, structure


def unpack_args_kwargs(flat_tensors: List[torch.Tensor], structure: Any):
    """Returns (args, kwargs)."""
    args, kwargs = _unpack(structure, flat_tensors)
    return list(args), kwargs
