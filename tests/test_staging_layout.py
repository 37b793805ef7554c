"""CPU mirror of the HIP pack kernel's gather math.

``py_pack`` reproduces, in Python, exactly the address computation the
pack kernel performs from a PackItem descriptor (row decomposition over
outer dims, innermost contiguous run). Its output must equal
``t.contiguous()``'s bytes for every layout — the same oracle the GPU
numerics tests use (plain torch fp32 reference)."""

import pytest
import torch

from torchsnapshot_amd.ops.staging import (
    ALIGN,
    build_pack_items,
    collapse_layout,
)


def py_pack(t: torch.Tensor) -> bytes:
    items, _, _ = build_pack_items([t])
    it = items[0]
    if it.nbytes == 0:
        return b""
    storage = t.untyped_storage()
    # read the storage bytes directly
    u8 = torch.empty(storage.nbytes(), dtype=torch.uint8)
    u8.untyped_storage().copy_(storage)
    storage_b = bytes(u8.numpy())
    base = t.data_ptr() - storage.data_ptr()
    rb = it.row_bytes
    out = bytearray(it.nbytes)
    rows = it.nbytes // rb
    for r in range(rows):
        off = 0
        rem = r
        for k in reversed(range(len(it.outer_sizes))):
            idx = rem % it.outer_sizes[k]
            rem //= it.outer_sizes[k]
            off += idx * it.outer_strides[k]
        src = base + off
        out[r * rb : (r + 1) * rb] = storage_b[src : src + rb]
    return bytes(out)


def ref_bytes(t: torch.Tensor) -> bytes:
    c = t.contiguous().reshape(-1)
    if c.numel() == 0:
        return b""
    return bytes(c.view(torch.uint8).numpy())


_CASES = [
    ("contig", lambda: torch.randn(64, 32)),
    ("transpose", lambda: torch.randn(32, 48).t()),
    ("permute3d", lambda: torch.randn(8, 9, 10).permute(2, 0, 1)),
    ("narrow0", lambda: torch.randn(100, 7)[20:60]),
    ("narrow1", lambda: torch.randn(50, 40)[:, 8:24]),
    ("strided", lambda: torch.randn(61, 63)[::3, ::2]),
    ("expanded", lambda: torch.randn(1, 16).expand(8, 16)),
    ("scalar", lambda: torch.tensor(3.5)),
    ("flipped-ish", lambda: torch.randn(6, 5, 4, 3).permute(3, 1, 0, 2)),
    ("size1dims", lambda: torch.randn(4, 1, 8, 1)),
    ("bf16", lambda: torch.randn(33, 17).to(torch.bfloat16).t()),
    ("u8", lambda: (torch.rand(129, 7) * 255).to(torch.uint8)),
]


@pytest.mark.parametrize("name,make", _CASES, ids=[c[0] for c in _CASES])
def test_pack_math_matches_contiguous(name, make):
    torch.manual_seed(0)
    t = make()
    assert py_pack(t) == ref_bytes(t)


def test_collapse_layout_logical_order():
    # collapsed (sizes, strides) must describe the tensor's LOGICAL order:
    # as_strided over the same storage equals the original reshaped
    for name, make in _CASES:
        t = make()
        sizes, strides = collapse_layout(t)
        v = torch.as_strided(t, sizes, strides, t.storage_offset())
        assert torch.equal(
            v.contiguous().reshape(-1), t.contiguous().reshape(-1)
        ), name


def test_batch_offsets_aligned():
    tensors = [torch.randn(13, 7), torch.randn(5), torch.randn(3, 3)]
    items, offsets, total = build_pack_items(tensors)
    for off in offsets:
        assert off % ALIGN == 0
    assert total >= sum(t.numel() * 4 for t in tensors)


def test_vec_divides_everything():
    t = torch.randn(64, 32).t()
    items, _, _ = build_pack_items([t])
    it = items[0]
    assert it.row_bytes % it.vec == 0
    for st in it.outer_strides:
        if st:
            assert st % it.vec == 0
