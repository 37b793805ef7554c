"""Tensor (de)serialization.

Two serializers (parity with reference torchsnapshot/serialization.py):

- ``buffer``: the raw bytes of a contiguous tensor, exposed zero-copy as a
  memoryview (bfloat16/fp8/complex included, via a uint8 reinterpret view).
  This is the fast path: no pickling, and on write the memoryview is handed
  straight to the storage layer.
- ``torch_save``: ``torch.save`` bytes, used for dtypes without a stable raw
  layout across builds (quantized tensors) and for arbitrary objects.
"""

from __future__ import annotations

import io
from typing import Tuple

import torch

# -- dtype tables ------------------------------------------------------------

_ALL_DTYPES = [
    torch.float32,
    torch.float64,
    torch.float16,
    torch.bfloat16,
    torch.complex64,
    torch.complex128,
    torch.uint8,
    torch.int8,
    torch.int16,
    torch.int32,
    torch.int64,
    torch.bool,
]
# Newer dtypes, present on this torch build but guarded for safety.
for _name in (
    "float8_e4m3fn",
    "float8_e5m2",
    "float8_e4m3fnuz",
    "float8_e5m2fnuz",
    "uint16",
    "uint32",
    "uint64",
):
    _dt = getattr(torch, _name, None)
    if _dt is not None:
        _ALL_DTYPES.append(_dt)

_QUANTIZED_DTYPES = [torch.qint8, torch.quint8, torch.qint32]


def dtype_to_str(dtype: torch.dtype) -> str:
    return str(dtype).removeprefix("torch.")


STR_TO_DTYPE = {dtype_to_str(d): d for d in _ALL_DTYPES + _QUANTIZED_DTYPES}
DTYPE_TO_STR = {d: s for s, d in STR_TO_DTYPE.items()}


def str_to_dtype(s: str) -> torch.dtype:
    try:
        return STR_TO_DTYPE[s]
    except KeyError:
        raise ValueError(f"unknown dtype string in manifest: {s!r}") from None


def dtype_size_bytes(dtype_str: str) -> int:
    dtype = str_to_dtype(dtype_str)
    if dtype.is_floating_point or dtype.is_complex:
        return dtype.itemsize
    return dtype.itemsize


SERIALIZER_BUFFER = "buffer"
SERIALIZER_TORCH_SAVE = "torch_save"


def pick_serializer(tensor: torch.Tensor) -> str:
    """Raw-buffer serialization for every plain dtype; torch_save for
    quantized tensors (their scale/zero-point layout is torch-internal)."""
    if tensor.is_quantized:
        return SERIALIZER_TORCH_SAVE
    if tensor.dtype in DTYPE_TO_STR:
        return SERIALIZER_BUFFER
    return SERIALIZER_TORCH_SAVE


# -- buffer serializer -------------------------------------------------------


def tensor_as_memoryview(tensor: torch.Tensor) -> memoryview:
    """Zero-copy view of a CPU tensor's bytes.

    The tensor must be contiguous (callers materialize views first so we
    never serialize bytes the tensor doesn't logically own). Works for all
    non-quantized dtypes including bfloat16/fp8 via a uint8 reinterpret.
    """
    if tensor.device.type != "cpu":
        raise ValueError("tensor_as_memoryview requires a CPU tensor")
    if tensor.is_quantized:
        raise ValueError("quantized tensors use the torch_save serializer")
    if not tensor.is_contiguous():
        raise ValueError("tensor_as_memoryview requires a contiguous tensor")
    flat = tensor.reshape(-1)
    if flat.numel() == 0:
        return memoryview(b"")
    u8 = flat.view(torch.uint8)
    return memoryview(u8.numpy())  # shares memory with the tensor


def tensor_from_memoryview(
    mv: memoryview, dtype: torch.dtype, shape: Tuple[int, ...]
) -> torch.Tensor:
    """Zero-copy: wrap a buffer as a tensor of the given dtype/shape."""
    numel = 1
    for s in shape:
        numel *= s
    if numel == 0:
        return torch.empty(shape, dtype=dtype)
    u8 = torch.frombuffer(mv, dtype=torch.uint8)
    return u8.view(dtype).reshape(shape)


# -- torch_save serializer ---------------------------------------------------


def torch_save_as_bytes(obj: object) -> bytes:
    buf = io.BytesIO()
    torch.save(obj, buf)
    return buf.getvalue()


def torch_load_from_bytes(data: bytes) -> object:
    return torch.load(io.BytesIO(data), weights_only=False)
