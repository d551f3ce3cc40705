"""Small shared helpers (parity: reference utils/misc.py:3-11 DUMMY sentinel)."""

from __future__ import annotations

import torch

DUMMY = torch.empty(0)  # dummy tensor that replaces empty prompt or adapter parameters
DUMMY_INT64 = torch.empty(0, dtype=torch.int64)


def is_dummy(tensor: torch.Tensor) -> bool:
    return tensor.numel() == 0


DTYPE_BYTES = {
    torch.float32: 4,
    torch.float16: 2,
    torch.bfloat16: 2,
    torch.int8: 1,
    torch.uint8: 1,
    torch.int64: 8,
    torch.bool: 1,
}


def get_size_in_bytes(dtype: torch.dtype) -> int:
    return DTYPE_BYTES[dtype]


def docstring_from(source):
    def wrapper(fn):
        fn.__doc__ = source.__doc__
        return fn

    return wrapper


def sample_up_to(population, k: int):
    """Up to k random items (parity: reference utils/random.py:7)."""
    import random

    population = list(population)
    if len(population) > k:
        population = random.sample(population, k)
    return population
