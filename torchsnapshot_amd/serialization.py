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
SERIALIZER_QTENSOR = "qtensor"


def pick_serializer(tensor: torch.Tensor) -> str:
    """Raw-buffer serialization for every plain dtype; a compact custom
    binary layout for quantized tensors (parity with reference
    torchsnapshot/serialization.py:278-477); torch_save for anything
    exotic."""
    if tensor.is_quantized:
        if tensor.qscheme() in (
            torch.per_tensor_affine,
            torch.per_channel_affine,
        ):
            return SERIALIZER_QTENSOR
        return SERIALIZER_TORCH_SAVE
    if tensor.layout != torch.strided:
        # sparse (COO/CSR/...) tensors have no flat storage to serialize
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
    if flat.stride(0) != 1:
        # size-1 dims make is_contiguous() true for ANY stride (e.g. a
        # step-2 slice of a 2-element tensor -> shape (1,), stride (2,)),
        # but the uint8 reinterpret needs a unit stride; re-materialize
        # (trivial: such tensors have one element per weird dim)
        flat = torch.as_strided(flat.clone(), flat.shape, (1,))
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


# -- quantized-tensor serializer ---------------------------------------------
#
# Layout (little-endian):
#   u32 header_len | header json | raw storage bytes | scales f64[] | zps i64[]
# Per-tensor: one scale + one zero point. Per-channel: axis in the header,
# C scales + C zero points. The storage bytes are the int representation
# (int_repr), so the payload stays zero-copy-readable.


def qtensor_as_bytes(tensor: torch.Tensor) -> bytes:
    import json as _json
    import struct as _struct

    if tensor.qscheme() == torch.per_tensor_affine:
        header = {
            "scheme": "per_tensor",
            "dtype": dtype_to_str(tensor.dtype),
            "shape": list(tensor.shape),
        }
        scales = _struct.pack("<d", float(tensor.q_scale()))
        zps = _struct.pack("<q", int(tensor.q_zero_point()))
    elif tensor.qscheme() == torch.per_channel_affine:
        header = {
            "scheme": "per_channel",
            "dtype": dtype_to_str(tensor.dtype),
            "shape": list(tensor.shape),
            "axis": int(tensor.q_per_channel_axis()),
        }
        s = tensor.q_per_channel_scales().to(torch.float64).contiguous()
        z = tensor.q_per_channel_zero_points().to(torch.int64).contiguous()
        scales = bytes(tensor_as_memoryview(s))
        zps = bytes(tensor_as_memoryview(z))
    else:
        raise ValueError(f"unsupported qscheme: {tensor.qscheme()}")
    hdr = _json.dumps(header).encode("utf-8")
    storage = bytes(tensor_as_memoryview(tensor.int_repr().contiguous()))
    return (
        _struct.pack("<I", len(hdr)) + hdr + storage + scales + zps
    )


def qtensor_from_bytes(data) -> torch.Tensor:
    import json as _json
    import struct as _struct

    mv = memoryview(data)
    (hdr_len,) = _struct.unpack("<I", mv[:4])
    header = _json.loads(bytes(mv[4 : 4 + hdr_len]).decode("utf-8"))
    dtype = str_to_dtype(header["dtype"])
    shape = tuple(header["shape"])
    numel = 1
    for s in shape:
        numel *= s
    # int_repr element size: qint8/quint8 -> 1, qint32 -> 4
    int_dtype = {
        torch.qint8: torch.int8,
        torch.quint8: torch.uint8,
        torch.qint32: torch.int32,
    }[dtype]
    storage_len = numel * int_dtype.itemsize
    off = 4 + hdr_len
    storage = tensor_from_memoryview(
        mv[off : off + storage_len], int_dtype, shape
    )
    off += storage_len
    if header["scheme"] == "per_tensor":
        (scale,) = _struct.unpack("<d", mv[off : off + 8])
        (zp,) = _struct.unpack("<q", mv[off + 8 : off + 16])
        return torch._make_per_tensor_quantized_tensor(storage, scale, zp)
    n_ch = shape[header["axis"]]
    scales = tensor_from_memoryview(
        mv[off : off + 8 * n_ch], torch.float64, (n_ch,)
    )
    off += 8 * n_ch
    zps = tensor_from_memoryview(mv[off : off + 8 * n_ch], torch.int64, (n_ch,))
    return torch._make_per_channel_quantized_tensor(
        storage, scales, zps, header["axis"]
    )


# -- torch_save serializer ---------------------------------------------------


def torch_save_as_bytes(obj: object) -> bytes:
    buf = io.BytesIO()
    torch.save(obj, buf)
    return buf.getvalue()


def torch_load_from_bytes(data: bytes) -> object:
    return torch.load(io.BytesIO(data), weights_only=False)
