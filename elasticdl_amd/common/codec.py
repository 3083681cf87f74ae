"""Binary message codec for the control and PS data planes.

The reference ships tensors as TensorFlow `TensorProto`s over gRPC
(elasticdl/proto/elasticdl.proto:47-86, common/tensor_utils.py:25-122).
This rebuild drops protobuf entirely: a message is a msgpack-encoded
structure in which tensors are replaced by blob references, followed by the
raw little-endian tensor bytes. Decoding a tensor is a zero-copy
``torch.frombuffer`` view over the message buffer — no per-element parsing,
no protobuf allocation, which matters at PS data-plane rates (whole dense
models and [n, dim] embedding batches per message).

Wire format::

    [u32 header_len][msgpack header][blob 0][blob 1]...

The header is any msgpack-serializable structure (dicts/lists/str/int/...)
where each tensor has been replaced by
``{"__tensor__": blob_index, "dtype": str, "shape": [..]}``. Blobs are
8-byte aligned so frombuffer views are aligned for every dtype.
"""

import struct
import warnings
from typing import Any, List, Tuple

import msgpack
import numpy as np
import torch

_HEADER_LEN = struct.Struct("<I")
_ALIGN = 8

# dtype registry: wire name <-> torch dtype <-> numpy dtype (for CPU paths)
_TORCH_TO_NAME = {
    torch.float32: "f32",
    torch.float64: "f64",
    torch.float16: "f16",
    torch.bfloat16: "bf16",
    torch.int64: "i64",
    torch.int32: "i32",
    torch.int16: "i16",
    torch.int8: "i8",
    torch.uint8: "u8",
    torch.bool: "b1",
}
_NAME_TO_TORCH = {v: k for k, v in _TORCH_TO_NAME.items()}
_NAME_TO_NP = {
    "f32": np.float32,
    "f64": np.float64,
    "f16": np.float16,
    # bf16 has no numpy dtype: views use uint16 storage
    "bf16": np.uint16,
    "i64": np.int64,
    "i32": np.int32,
    "i16": np.int16,
    "i8": np.int8,
    "u8": np.uint8,
    "b1": np.bool_,
}


def dtype_name(dtype: torch.dtype) -> str:
    return _TORCH_TO_NAME[dtype]


def name_to_dtype(name: str) -> torch.dtype:
    return _NAME_TO_TORCH[name]


def _pack_structure(obj: Any, blobs: List[bytes]) -> Any:
    if isinstance(obj, torch.Tensor):
        t = obj.detach()
        if t.device.type != "cpu":
            t = t.cpu()
        t = t.contiguous()
        idx = len(blobs)
        # bf16 and other non-numpy dtypes: view storage as uint8
        blobs.append(t.view(torch.uint8).numpy().tobytes()
                     if t.dtype == torch.bfloat16
                     else t.numpy().tobytes())
        return {
            "__tensor__": idx,
            "dtype": _TORCH_TO_NAME[t.dtype],
            "shape": list(t.shape),
        }
    if isinstance(obj, np.ndarray):
        return _pack_structure(torch.from_numpy(np.ascontiguousarray(obj)), blobs)
    if isinstance(obj, dict):
        return {k: _pack_structure(v, blobs) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [_pack_structure(v, blobs) for v in obj]
    return obj


def encode(message: Any) -> bytes:
    """Encode a message structure (with embedded torch tensors) to bytes."""
    blobs: List[bytes] = []
    structure = _pack_structure(message, blobs)
    # compute aligned offsets after the header; header itself records offsets
    sizes = [len(b) for b in blobs]
    header = {"s": structure, "n": len(blobs), "sz": sizes}
    header_bytes = msgpack.packb(header, use_bin_type=True)
    parts = [_HEADER_LEN.pack(len(header_bytes)), header_bytes]
    pos = _HEADER_LEN.size + len(header_bytes)
    for b in blobs:
        pad = (-pos) % _ALIGN
        if pad:
            parts.append(b"\x00" * pad)
            pos += pad
        parts.append(b)
        pos += len(b)
    return b"".join(parts)


def _unpack_structure(obj: Any, blob_views: List[Tuple[int, int]], buf: memoryview) -> Any:
    if isinstance(obj, dict):
        if "__tensor__" in obj:
            start, size = blob_views[obj["__tensor__"]]
            name = obj["dtype"]
            shape = obj["shape"]
            raw = np.frombuffer(buf, dtype=np.uint8, count=size, offset=start)
            with warnings.catch_warnings():
                # decoded tensors are intentionally read-only zero-copy views
                warnings.simplefilter("ignore", UserWarning)
                t = torch.from_numpy(raw)
            dt = _NAME_TO_TORCH[name]
            t = t.view(dt)
            return t.reshape(shape)
        return {k: _unpack_structure(v, blob_views, buf) for k, v in obj.items()}
    if isinstance(obj, list):
        return [_unpack_structure(v, blob_views, buf) for v in obj]
    return obj


def decode(data: bytes) -> Any:
    """Decode bytes back to the message structure; tensors are zero-copy
    views over ``data`` (callers must .clone() before in-place mutation)."""
    buf = memoryview(data)
    (hlen,) = _HEADER_LEN.unpack_from(buf, 0)
    header = msgpack.unpackb(bytes(buf[_HEADER_LEN.size:_HEADER_LEN.size + hlen]), raw=False)
    pos = _HEADER_LEN.size + hlen
    blob_views: List[Tuple[int, int]] = []
    for size in header["sz"]:
        pos += (-pos) % _ALIGN
        blob_views.append((pos, size))
        pos += size
    return _unpack_structure(header["s"], blob_views, buf)
