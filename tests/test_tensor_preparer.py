"""Tensor preparer unit tests without storage: fulfill stager buffers
directly into consumers (the reference's preparer-test pattern)."""

import asyncio
from concurrent.futures import ThreadPoolExecutor

import pytest
import torch

from torchsnapshot_amd.io_preparer import prepare_read, prepare_write
from torchsnapshot_amd.io_types import StageContext
from torchsnapshot_amd.test_utils import rand_tensor, tensor_eq


def _run(coro):
    return asyncio.run(coro)


async def _stage(req, is_async=False):
    ctx = StageContext(executor=ThreadPoolExecutor(2), is_async=is_async)
    return await req.stager.stage_buffer(ctx)


async def _consume(req, buf):
    ctx = StageContext(executor=ThreadPoolExecutor(2))
    await req.consumer.consume_buffer(ctx, buf)
    req.consumer.close()


def _round_trip(tensor, obj_out=None, is_async=False):
    entry, write_reqs = prepare_write(tensor, "p", rank=0, is_async_snapshot=is_async)
    bufs = [_run(_stage(wr, is_async)) for wr in write_reqs]
    read_reqs, fut = prepare_read(entry, obj_out)
    assert len(read_reqs) == len(bufs)
    for rr, buf in zip(read_reqs, bufs):
        _run(_consume(rr, bytearray(memoryview(buf))))
    for wr in write_reqs:
        wr.stager.release_buffer()
    return fut.obj


@pytest.mark.parametrize(
    "dtype",
    [torch.float32, torch.bfloat16, torch.int64, torch.bool, torch.qint8],
    ids=str,
)
def test_round_trip_dtypes(dtype):
    t = rand_tensor((9, 5), dtype)
    out = _round_trip(t)
    assert tensor_eq(t, out)


def test_round_trip_into_strided_dst():
    t = torch.rand(6, 6)
    base = torch.zeros(12, 12)
    dst = base[::2, ::2]
    out = _round_trip(t, obj_out=dst)
    assert out is dst
    assert torch.equal(base[::2, ::2], t)


def test_sync_stage_is_zero_copy():
    """Sync staging of a plain CPU tensor shares memory: mutations BEFORE
    the write hits storage would be visible (this is why sync take only
    returns after I/O completes)."""
    t = torch.zeros(64)
    entry, (wr,) = prepare_write(t, "p", rank=0, is_async_snapshot=False)
    buf = _run(_stage(wr))
    t.fill_(7.0)
    assert torch.frombuffer(bytearray(memoryview(buf)), dtype=torch.float32)[0] == 7.0


def test_async_stage_is_a_copy():
    """Async staging must defensively copy CPU tensors: training mutates
    them while storage I/O is still draining."""
    t = torch.zeros(64)
    entry, (wr,) = prepare_write(t, "p", rank=0, is_async_snapshot=True)
    buf = _run(_stage(wr, is_async=True))
    t.fill_(7.0)
    assert torch.frombuffer(bytearray(memoryview(buf)), dtype=torch.float32)[0] == 0.0


def test_view_tensor_staged_as_logical_content():
    base = torch.rand(10, 10)
    view = base[2:4]
    entry, (wr,) = prepare_write(view, "p", rank=0)
    buf = _run(_stage(wr))
    expect = bytes(view.contiguous().reshape(-1).view(torch.uint8).numpy())
    assert bytes(memoryview(buf)) == expect
    # view over a larger storage must have been copied (no byte leak)
    assert memoryview(buf).nbytes == view.numel() * 4


def test_chunked_entry_round_trip():
    from torchsnapshot_amd import knobs

    with knobs.override_max_chunk_size_bytes(256):
        t = torch.rand(64, 4)
        entry, write_reqs = prepare_write(t, "p", rank=0)
        assert len(write_reqs) > 1
        bufs = {wr.path: _run(_stage(wr)) for wr in write_reqs}
        read_reqs, fut = prepare_read(entry, torch.zeros(64, 4))
        for rr in read_reqs:
            _run(_consume(rr, bytearray(memoryview(bufs[rr.path]))))
        assert torch.equal(fut.obj, t)


def test_primitive_no_write_reqs():
    entry, write_reqs = prepare_write(42, "p", rank=0)
    assert write_reqs == []
    read_reqs, fut = prepare_read(entry)
    assert read_reqs == []
    assert fut.obj == 42
