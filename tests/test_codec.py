import numpy as np
import pytest
import torch

from elasticdl_amd.common import codec


def roundtrip(msg):
    return codec.decode(codec.encode(msg))


def test_scalar_structures():
    msg = {"a": 1, "b": "x", "c": [1, 2, {"d": None}], "e": 3.5, "f": True}
    assert roundtrip(msg) == msg


@pytest.mark.parametrize(
    "dtype",
    [torch.float32, torch.float64, torch.float16, torch.bfloat16,
     torch.int64, torch.int32, torch.uint8, torch.bool],
)
def test_tensor_roundtrip(dtype):
    if dtype == torch.bool:
        t = torch.rand(7, 3) > 0.5
    elif dtype.is_floating_point:
        t = torch.randn(7, 3).to(dtype)
    else:
        t = torch.randint(0, 100, (7, 3), dtype=dtype)
    out = roundtrip({"t": t})["t"]
    assert out.dtype == dtype
    assert out.shape == t.shape
    assert torch.equal(out, t)


def test_nested_tensors_and_alignment():
    msg = {
        "dense": {f"p{i}": torch.randn(5, i + 1) for i in range(4)},
        "ids": torch.arange(11, dtype=torch.int64),
        "meta": {"version": 7},
        "odd": torch.arange(3, dtype=torch.uint8),  # 3-byte blob forces padding
        "after": torch.randn(2, 2, dtype=torch.float64),
    }
    out = roundtrip(msg)
    assert out["meta"]["version"] == 7
    for k, v in msg["dense"].items():
        assert torch.equal(out["dense"][k], v)
    assert torch.equal(out["ids"], msg["ids"])
    assert torch.equal(out["after"], msg["after"])


def test_numpy_input():
    a = np.random.randn(4, 4).astype(np.float32)
    out = roundtrip({"a": a})["a"]
    assert torch.equal(out, torch.from_numpy(a))


def test_empty_tensor():
    out = roundtrip({"t": torch.empty(0, 8)})["t"]
    assert out.shape == (0, 8)


def test_zero_copy_view():
    t = torch.randn(1024)
    data = codec.encode({"t": t})
    out = codec.decode(data)["t"]
    assert torch.equal(out, t)
