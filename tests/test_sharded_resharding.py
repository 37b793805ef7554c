"""Resharding unit tests (no storage, no dist): fulfill write-request
buffers directly into read-request consumers, across sharding changes —
the reference's preparer-level test pattern
(tests/test_sharded_tensor_resharding.py)."""

import asyncio
from concurrent.futures import ThreadPoolExecutor

import pytest
import torch

from torchsnapshot_amd.io_types import StageContext
from torchsnapshot_amd.io_preparers.sharded_tensor import (
    compute_overlap,
    narrow_nd,
    plan_shard_reads,
    subdivide_shard,
)
from torchsnapshot_amd.manifest import Shard as ShardMeta
from torchsnapshot_amd.manifest import TensorEntry
from torchsnapshot_amd.serialization import dtype_to_str, tensor_as_memoryview


def test_compute_overlap_basic():
    ov = compute_overlap([0, 0], [4, 4], [2, 2], [4, 4])
    assert ov.src_offsets == [2, 2]
    assert ov.dst_offsets == [0, 0]
    assert ov.lengths == [2, 2]


def test_compute_overlap_none():
    assert compute_overlap([0], [4], [4], [4]) is None
    assert compute_overlap([0, 0], [4, 4], [0, 4], [4, 4]) is None


def test_narrow_nd():
    t = torch.arange(24).reshape(4, 6)
    v = narrow_nd(t, [1, 2], [2, 3])
    assert torch.equal(v, t[1:3, 2:5])


def test_subdivide_shard():
    t = torch.rand(100, 10)  # 4000 bytes/row... 100*10*4 = 4000 B total
    pieces = subdivide_shard(t, [50, 0], max_bytes=1000)
    assert len(pieces) == 4
    total = 0
    for piece, offsets in pieces:
        assert offsets[1] == 0
        total += piece.shape[0]
    assert total == 100
    # offsets accumulate along dim 0 from the global base 50
    assert pieces[0][1][0] == 50
    assert pieces[-1][1][0] == 50 + 100 - pieces[-1][0].shape[0]


def _simulate(shards_spec, target_views):
    """shards_spec: [(offsets, tensor)] persisted; target_views:
    [(view_tensor, view_offsets)]. Runs consumers with buffers produced
    straight from the persisted tensors."""
    shards_meta = []
    payloads = {}
    for offsets, tensor in shards_spec:
        loc = f"shard_{'_'.join(map(str, offsets))}"
        entry = TensorEntry(
            location=loc,
            serializer="buffer",
            dtype=dtype_to_str(tensor.dtype),
            shape=list(tensor.shape),
        )
        shards_meta.append(
            ShardMeta(offsets=list(offsets), sizes=list(tensor.shape), tensor=entry)
        )
        payloads[loc] = bytearray(tensor_as_memoryview(tensor.contiguous()))
    read_reqs = plan_shard_reads(shards_meta, target_views)

    async def run():
        ctx = StageContext(executor=ThreadPoolExecutor(2))
        for rr in read_reqs:
            await rr.consumer.consume_buffer(ctx, memoryview(payloads[rr.path]))

    asyncio.run(run())


@pytest.mark.parametrize("src_parts,dst_parts", [(2, 4), (4, 2), (3, 5), (1, 7)])
def test_reshard_dim0(src_parts, dst_parts):
    full = torch.rand(60, 8)
    # persist split into src_parts along dim0
    shards = []
    step = 60 // src_parts
    for i in range(src_parts):
        lo = i * step
        hi = 60 if i == src_parts - 1 else (i + 1) * step
        shards.append(([lo, 0], full[lo:hi].clone()))
    # restore into dst_parts views
    out = torch.zeros(60, 8)
    views = []
    step = 60 // dst_parts
    for i in range(dst_parts):
        lo = i * step
        hi = 60 if i == dst_parts - 1 else (i + 1) * step
        views.append((out[lo:hi], [lo, 0]))
    _simulate(shards, views)
    assert torch.equal(out, full)


def test_reshard_dim0_to_dim1():
    full = torch.rand(16, 16)
    shards = [([0, 0], full[:8].clone()), ([8, 0], full[8:].clone())]
    out = torch.zeros(16, 16)
    views = [(out[:, :4], [0, 0]), (out[:, 4:], [0, 4])]
    _simulate(shards, views)
    assert torch.equal(out, full)


def test_reshard_2d_grid():
    full = torch.rand(12, 12)
    shards = []
    for i in range(3):
        for j in range(2):
            shards.append(
                ([i * 4, j * 6], full[i * 4 : (i + 1) * 4, j * 6 : (j + 1) * 6].clone())
            )
    out = torch.zeros(12, 12)
    views = [(out[k * 2 : (k + 1) * 2, :], [k * 2, 0]) for k in range(6)]
    _simulate(shards, views)
    assert torch.equal(out, full)


def test_load_into_full_tensor():
    full = torch.rand(10, 4)
    shards = [([0, 0], full[:5].clone()), ([5, 0], full[5:].clone())]
    out = torch.zeros(10, 4)
    _simulate(shards, [(out, [0, 0])])
    assert torch.equal(out, full)


def _row_views(out: torch.Tensor, parts: int):
    """torch.chunk-style (ceil, possibly-empty-tail) row split of ``out``."""
    n = out.shape[0]
    step = -(-n // parts)
    views = []
    for i in range(parts):
        lo = min(i * step, n)
        hi = min(lo + step, n)
        if hi > lo:
            views.append((out[lo:hi], [lo] + [0] * (out.dim() - 1)))
    return views


@pytest.mark.parametrize("dim", [0, 1])
@pytest.mark.parametrize("src_parts", [1, 2, 3, 5])
@pytest.mark.parametrize("dst_parts", [1, 2, 4, 7])
def test_reshard_matrix_dims_counts(dim, src_parts, dst_parts):
    """The reference's ChunkShardingSpec matrix: every (shard dim, source
    count, dest count) combination, sizes chosen indivisible (47, 23)
    (reference tests/test_sharded_tensor_resharding.py:98-110)."""
    full = torch.rand(47, 23)
    n = full.shape[dim]

    def split(parts):
        step = -(-n // parts)
        pieces = []
        for i in range(parts):
            lo = min(i * step, n)
            hi = min(lo + step, n)
            if hi > lo:
                off = [0, 0]
                off[dim] = lo
                pieces.append((off, full.narrow(dim, lo, hi - lo).clone()))
        return pieces

    shards = split(src_parts)
    out = torch.zeros(47, 23)
    views = []
    step = -(-n // dst_parts)
    for i in range(dst_parts):
        lo = min(i * step, n)
        hi = min(lo + step, n)
        if hi > lo:
            off = [0, 0]
            off[dim] = lo
            views.append((out.narrow(dim, lo, hi - lo), off))
    _simulate(shards, views)
    assert torch.equal(out, full)


def test_reshard_enumerable_irregular():
    """EnumerableShardingSpec-style irregular tiling: explicitly-placed
    unequal rectangles covering the tensor, restored into row-wise views
    and into a 2-D grid."""
    full = torch.rand(10, 12)
    shards = [
        ([0, 0], full[0:3, 0:12].clone()),          # wide top strip
        ([3, 0], full[3:10, 0:5].clone()),          # tall left block
        ([3, 5], full[3:6, 5:12].clone()),          # mid right block
        ([6, 5], full[6:10, 5:9].clone()),          # lower middle
        ([6, 9], full[6:10, 9:12].clone()),         # lower right
    ]
    out = torch.zeros(10, 12)
    _simulate(shards, _row_views(out, 4))
    assert torch.equal(out, full)

    out2 = torch.zeros(10, 12)
    grid_views = [
        (out2[0:5, 0:6], [0, 0]),
        (out2[0:5, 6:12], [0, 6]),
        (out2[5:10, 0:6], [5, 0]),
        (out2[5:10, 6:12], [5, 6]),
    ]
    _simulate(shards, grid_views)
    assert torch.equal(out2, full)


def test_reshard_random_tilings():
    """Randomized rectangular tilings on both sides (seeded): recursively
    split the tensor into irregular tiles, persist one tiling, restore
    into another."""
    g = torch.Generator().manual_seed(1234)

    def tile(lo0, hi0, lo1, hi1, depth):
        if depth == 0 or (hi0 - lo0 < 2 and hi1 - lo1 < 2):
            return [(lo0, hi0, lo1, hi1)]
        if (hi0 - lo0 >= 2) and (
            hi1 - lo1 < 2 or int(torch.randint(0, 2, (1,), generator=g)) == 0
        ):
            cut = lo0 + 1 + int(
                torch.randint(0, hi0 - lo0 - 1, (1,), generator=g)
            )
            return tile(lo0, cut, lo1, hi1, depth - 1) + tile(
                cut, hi0, lo1, hi1, depth - 1
            )
        cut = lo1 + 1 + int(torch.randint(0, hi1 - lo1 - 1, (1,), generator=g))
        return tile(lo0, hi0, lo1, cut, depth - 1) + tile(
            lo0, hi0, cut, hi1, depth - 1
        )

    for trial in range(4):
        full = torch.rand(21, 17)
        src = tile(0, 21, 0, 17, 3)
        dst = tile(0, 21, 0, 17, 3)
        shards = [
            ([a, c], full[a:b, c:d].clone()) for a, b, c, d in src
        ]
        out = torch.zeros(21, 17)
        views = [(out[a:b, c:d], [a, c]) for a, b, c, d in dst]
        _simulate(shards, views)
        assert torch.equal(out, full), f"trial {trial}"
