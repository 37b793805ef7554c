import pytest
import torch

from torchsnapshot_amd.serialization import (
    SERIALIZER_BUFFER,
    dtype_to_str,
    pick_serializer,
    str_to_dtype,
    tensor_as_memoryview,
    tensor_from_memoryview,
    torch_load_from_bytes,
    torch_save_as_bytes,
)
from torchsnapshot_amd.test_utils import rand_tensor, tensor_eq

_PLAIN_DTYPES = [
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
    torch.float8_e4m3fn,
    torch.float8_e5m2,
]


@pytest.mark.parametrize("dtype", _PLAIN_DTYPES, ids=str)
def test_buffer_round_trip(dtype):
    t = rand_tensor((16, 9), dtype=dtype)
    mv = tensor_as_memoryview(t)
    assert mv.nbytes == t.numel() * t.element_size()
    # writable copy simulates a read buffer
    buf = bytearray(mv)
    t2 = tensor_from_memoryview(memoryview(buf), dtype=dtype, shape=(16, 9))
    assert tensor_eq(t, t2)


def test_dtype_string_round_trip():
    for dtype in _PLAIN_DTYPES + [torch.qint8, torch.quint8, torch.qint32]:
        assert str_to_dtype(dtype_to_str(dtype)) == dtype


def test_pick_serializer():
    from torchsnapshot_amd.serialization import SERIALIZER_QTENSOR

    assert pick_serializer(torch.rand(3)) == SERIALIZER_BUFFER
    assert pick_serializer(rand_tensor((3,), torch.qint8)) == SERIALIZER_QTENSOR


def test_scalar_tensor():
    t = torch.tensor(4.25, dtype=torch.bfloat16)
    mv = tensor_as_memoryview(t)
    t2 = tensor_from_memoryview(memoryview(bytearray(mv)), torch.bfloat16, ())
    assert t2.item() == 4.25


def test_empty_tensor():
    t = torch.empty(0, 5)
    mv = tensor_as_memoryview(t)
    assert mv.nbytes == 0
    t2 = tensor_from_memoryview(memoryview(b""), torch.float32, (0, 5))
    assert t2.shape == (0, 5)


def test_non_contiguous_rejected():
    t = torch.rand(4, 4).t()
    with pytest.raises(ValueError):
        tensor_as_memoryview(t)


def test_torch_save_round_trip():
    obj = {"a": torch.rand(3), "b": [1, 2]}
    data = torch_save_as_bytes(obj)
    obj2 = torch_load_from_bytes(data)
    assert torch.equal(obj["a"], obj2["a"])
    assert obj2["b"] == [1, 2]


def test_quantized_torch_save_round_trip():
    t = rand_tensor((8, 8), torch.qint8)
    t2 = torch_load_from_bytes(torch_save_as_bytes(t))
    assert tensor_eq(t, t2)


def test_memoryview_zero_copy():
    t = torch.rand(128)
    mv = tensor_as_memoryview(t)
    t[0] = 42.0
    # zero-copy: mutation visible through the view
    t2 = tensor_from_memoryview(mv, torch.float32, (128,))
    assert t2[0].item() == 42.0


def test_qtensor_layout_round_trip():
    from torchsnapshot_amd.serialization import (
        qtensor_as_bytes,
        qtensor_from_bytes,
    )

    for dtype in (torch.qint8, torch.quint8, torch.qint32):
        t = torch.quantize_per_tensor(
            torch.rand(13, 7), scale=0.07, zero_point=2, dtype=dtype
        )
        t2 = qtensor_from_bytes(qtensor_as_bytes(t))
        assert tensor_eq(t, t2)
        assert t2.q_scale() == t.q_scale()
        assert t2.q_zero_point() == t.q_zero_point()

    # per-channel
    t = torch.quantize_per_channel(
        torch.rand(6, 10),
        scales=torch.rand(6) * 0.1 + 0.01,
        zero_points=torch.randint(0, 10, (6,)),
        axis=0,
        dtype=torch.qint8,
    )
    t2 = qtensor_from_bytes(qtensor_as_bytes(t))
    assert tensor_eq(t, t2)
    assert torch.equal(t2.q_per_channel_scales(), t.q_per_channel_scales())
    assert t2.q_per_channel_axis() == 0


def test_memoryview_of_size1_weird_stride():
    """A step-2 slice of a 2-element tensor has shape (1,), stride (2,):
    is_contiguous() is True (size-1 dims allow any stride) but a uint8
    reinterpret needs unit stride — found by the snapshot fuzzer
    (seed 313949245)."""
    import torch

    from torchsnapshot_amd.serialization import (
        tensor_as_memoryview,
        tensor_from_memoryview,
    )

    t = torch.tensor([7, 9], dtype=torch.int16)[::2]
    assert t.is_contiguous() and t.stride() == (2,)
    mv = tensor_as_memoryview(t)
    back = tensor_from_memoryview(mv, dtype=torch.int16, shape=(1,))
    assert back.item() == 7
