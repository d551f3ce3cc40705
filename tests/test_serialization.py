import pytest
import torch

from petals_amd.utils import serialization as ser
from petals_amd.utils.packaging import pack_args_kwargs, unpack_args_kwargs


@pytest.mark.parametrize(
    "dtype", [torch.float32, torch.float16, torch.bfloat16, torch.int64, torch.int8, torch.bool]
)
def test_roundtrip_none(dtype):
    if dtype == torch.bool:
        t = torch.rand(3, 5) > 0.5
    elif dtype.is_floating_point:
        t = torch.randn(3, 5).to(dtype)
    else:
        t = torch.randint(-4 if dtype != torch.int8 else -4, 100, (3, 5)).to(dtype)
    desc, buf = ser.serialize_tensor(t, ser.NONE)
    t2 = ser.deserialize_tensor(desc, buf)
    assert t2.dtype == t.dtype and t2.shape == t.shape
    assert torch.equal(t2, t)


def test_roundtrip_fp16_compression():
    t = torch.randn(17, 33)
    desc, buf = ser.serialize_tensor(t, ser.FLOAT16)
    t2 = ser.deserialize_tensor(desc, buf)
    assert t2.dtype == torch.float32
    assert torch.allclose(t, t2, atol=1e-2, rtol=1e-2)
    assert len(buf) == t.numel() * 2


def test_roundtrip_blockwise8():
    t = torch.randn(5000) * 3
    desc, buf = ser.serialize_tensor(t, ser.BLOCKWISE_8BIT)
    t2 = ser.deserialize_tensor(desc, buf)
    assert t2.shape == t.shape
    assert torch.allclose(t, t2, atol=0.1, rtol=0.05)


def test_requires_grad_preserved():
    t = torch.randn(4, requires_grad=True)
    desc, buf = ser.serialize_tensor(t)
    t2 = ser.deserialize_tensor(desc, buf)
    assert t2.requires_grad


def test_pack_unpack_roundtrip():
    a, b = torch.randn(2, 3), torch.randn(4)
    tensors, structure = pack_args_kwargs(a, [b, {"x": a}], scale=1.5, t=(a, "s"))
    assert len(tensors) == 2  # deduplicated
    args, kwargs = unpack_args_kwargs(tensors, structure)
    assert torch.equal(args[0], a)
    assert torch.equal(args[1][0], b)
    assert torch.equal(args[1][1]["x"], a)
    assert kwargs["scale"] == 1.5
    assert kwargs["t"][1] == "s" and isinstance(kwargs["t"], tuple)
