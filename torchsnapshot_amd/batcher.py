"""Slab batching: many small tensor writes -> few large sequential I/Os.

Write side: buffer-serialized tensor write requests below the slab
threshold are grouped by device and packed into ``batched/<uuid>`` slab
files. For device tensors the whole slab is produced by ONE gather-pack
HIP kernel launch + ONE SDMA D2H copy (ops/staging.py) — this replaces the
reference's per-member UntypedStorage copies + ByteTensor slab
(torchsnapshot/batcher.py:104-162). For CPU tensors members are memcpy'd
into a bytearray slab by executor threads. Each member's TensorEntry is
relocated in place to (slab location, byte_range).

Read side: byte-ranged reads against the same file are merged into
spanning reads; a batched consumer slices the span and feeds each member
consumer (parity with torchsnapshot/batcher.py:358-478).
"""

from __future__ import annotations

import asyncio
from collections import defaultdict
from typing import Dict, List, Sequence

import torch

from . import knobs
from .io_types import (
    BufferConsumer,
    BufferStager,
    BufferType,
    ReadReq,
    StageContext,
    WriteReq,
)
from .io_preparers.tensor import TensorBufferStager
from .ops.staging import ALIGN, build_pack_items, get_pinned_pool, get_staging_engine
from .serialization import SERIALIZER_BUFFER


def _is_batchable(req: WriteReq) -> bool:
    stager = req.stager
    return (
        req.tensor_entry is not None
        and isinstance(stager, TensorBufferStager)
        and stager.serializer == SERIALIZER_BUFFER
        and not stager.tensor.is_quantized
        and stager.get_staging_cost_bytes() < knobs.get_slab_size_threshold_bytes()
    )


def batch_write_requests(
    write_reqs: List[WriteReq], rank: int = 0
) -> List[WriteReq]:
    """Group batchable requests into slab writes; returns the new request
    list (member entries are relocated in place). Slab names are
    deterministic per rank within one snapshot, so re-saving to the same
    path overwrites instead of accumulating slabs."""
    if knobs.is_batching_disabled():
        return write_reqs
    out: List[WriteReq] = []
    groups: Dict[str, List[WriteReq]] = defaultdict(list)
    for req in write_reqs:
        if _is_batchable(req):
            groups[str(req.stager.tensor.device)].append(req)
        else:
            out.append(req)

    slab_limit = knobs.get_slab_size_threshold_bytes()
    seq = 0
    for device_str, members in groups.items():
        if len(members) == 1:
            out.extend(members)
            continue
        # fill slabs greedily up to the limit
        cur: List[WriteReq] = []
        cur_bytes = 0
        for req in members:
            nbytes = req.stager.tensor.numel() * req.stager.tensor.element_size()
            aligned = (nbytes + ALIGN - 1) // ALIGN * ALIGN
            if cur and cur_bytes + aligned > slab_limit:
                out.append(_make_slab(device_str, cur, rank, seq))
                seq += 1
                cur, cur_bytes = [], 0
            cur.append(req)
            cur_bytes += aligned
        if len(cur) == 1:
            out.append(cur[0])
        elif cur:
            out.append(_make_slab(device_str, cur, rank, seq))
            seq += 1
    return out


def _make_slab(
    device_str: str, members: List[WriteReq], rank: int, seq: int
) -> WriteReq:
    slab_path = f"batched/{rank}-{seq}"
    tensors = [m.stager.tensor.detach() for m in members]
    _, offsets, total = build_pack_items(tensors)
    for m, off in zip(members, offsets):
        entry = m.tensor_entry
        nbytes = m.stager.tensor.numel() * m.stager.tensor.element_size()
        entry.location = slab_path
        entry.byte_range = [off, off + nbytes]
    is_async = any(m.stager.is_async_snapshot for m in members)
    stager = BatchedBufferStager(
        tensors=tensors, total_bytes=total, is_async_snapshot=is_async
    )
    return WriteReq(path=slab_path, stager=stager)


class BatchedBufferStager(BufferStager):
    def __init__(
        self,
        tensors: Sequence[torch.Tensor],
        total_bytes: int,
        is_async_snapshot: bool,
    ) -> None:
        self.tensors = tensors
        self.total_bytes = total_bytes
        self.is_async_snapshot = is_async_snapshot
        self._staged_batch = None
        self._pinned_block = None
        self.precomputed_checksum = None
        # [(start, end, "psum64:<hex>")] per member, recorded so restore
        # can verify byte-range/merged-span reads of this slab
        self.member_checksums = None

    def get_staging_cost_bytes(self) -> int:
        return self.total_bytes

    async def stage_buffer(self, ctx: StageContext) -> BufferType:
        loop = asyncio.get_running_loop()
        device = self.tensors[0].device
        if device.type == "cuda":
            return await loop.run_in_executor(ctx.executor, self._stage_device)
        return await loop.run_in_executor(ctx.executor, self._stage_cpu)

    def _stage_device(self) -> BufferType:
        from . import integrity

        engine = get_staging_engine(self.tensors[0].device)
        ck = integrity.checksumming_enabled()
        batch = engine.stage(self.tensors, compute_checksums=ck)
        batch.wait()
        if ck and batch.checksums is not None:
            # padding is zeroed, so the slab-file checksum is the sum of
            # the per-member device checksums
            total = sum(batch.checksums) % (1 << 64)
            self.precomputed_checksum = "psum64:" + format(total, "016x")
            self.member_checksums = [
                (off, off + n, "psum64:" + format(v % (1 << 64), "016x"))
                for off, n, v in zip(
                    batch.offsets, batch.nbytes_list, batch.checksums
                )
            ]
        self._staged_batch = batch
        return batch.slab_memoryview()

    def _stage_cpu(self) -> BufferType:
        from . import integrity

        items, offsets, total = build_pack_items(self.tensors)
        slab = bytearray(total)
        mv = memoryview(slab)
        from .serialization import tensor_as_memoryview

        ck = integrity.checksumming_enabled()
        members = []
        for t, off in zip(self.tensors, offsets):
            nbytes = t.numel() * t.element_size()
            if nbytes == 0:
                continue
            src = tensor_as_memoryview(t if t.is_contiguous() else t.contiguous())
            mv[off : off + nbytes] = src
            if ck:
                members.append(
                    (
                        off,
                        off + nbytes,
                        integrity.psum64_hexdigest(src, word_base=off // 8),
                    )
                )
        if ck:
            self.member_checksums = members
            total_ck = sum(
                int(v[len("psum64:"):], 16) for _, _, v in members
            ) % (1 << 64)
            self.precomputed_checksum = "psum64:" + format(total_ck, "016x")
        return mv

    def release_buffer(self) -> None:
        if self._staged_batch is not None:
            self._staged_batch.release()
            self._staged_batch = None
        self.tensors = ()  # free shadow clones as soon as the write lands


# ---------------------------------------------------------------------------
# read batching
# ---------------------------------------------------------------------------

_MERGE_GAP_BYTES = 4 * 1024 * 1024


def batch_read_requests(read_reqs: List[ReadReq]) -> List[ReadReq]:
    """Merge byte-ranged reads on the same file into spanning reads."""
    if knobs.is_batching_disabled():
        return read_reqs
    out: List[ReadReq] = []
    by_path: Dict[str, List[ReadReq]] = defaultdict(list)
    for rr in read_reqs:
        if rr.byte_range is not None:
            by_path[rr.path].append(rr)
        else:
            out.append(rr)

    limit = knobs.get_slab_size_threshold_bytes() * 2
    for path, members in by_path.items():
        if len(members) == 1:
            out.extend(members)
            continue
        members.sort(key=lambda r: r.byte_range[0])
        span: List[ReadReq] = []
        span_start = span_end = 0
        for rr in members:
            s, e = rr.byte_range
            if span and (s - span_end > _MERGE_GAP_BYTES or e - span_start > limit):
                out.append(_make_span(path, span, span_start, span_end))
                span = []
            if not span:
                span_start = s
            span.append(rr)
            span_end = max(span_end, e)
        if span:
            out.append(_make_span(path, span, span_start, span_end))
    return out


def _make_span(
    path: str, members: List[ReadReq], start: int, end: int
) -> ReadReq:
    if len(members) == 1:
        return members[0]
    consumer = BatchedBufferConsumer(members=members, span_start=start)
    return ReadReq(
        path=path,
        byte_range=(start, end),
        consumer=consumer,
        # all-device spans read straight into pinned memory
        buf_alloc=(
            consumer.alloc_pinned_buffer
            if consumer._device_fast_path_target() is not None
            else None
        ),
    )


class BatchedBufferConsumer(BufferConsumer):
    def __init__(self, members: List[ReadReq], span_start: int) -> None:
        self.members = members
        self.span_start = span_start
        self._pinned_block = None
        self._pinned_nbytes = 0
        # (expected psum64 of the whole span, word_base) set by the read
        # scheduler; verified on-device right after the span's single H2D
        self.expected_psum = None

    def alloc_pinned_buffer(self, nbytes: int) -> memoryview:
        self._pinned_block = get_pinned_pool().acquire(max(nbytes, 1))
        self._pinned_nbytes = nbytes
        return memoryview(self._pinned_block.tensor.numpy())[:nbytes]

    def close(self) -> None:
        if self._pinned_block is not None:
            get_pinned_pool().release(self._pinned_block)
            self._pinned_block = None

    def get_consuming_cost_bytes(self) -> int:
        total = 0
        for m in self.members:
            total += m.consumer.get_consuming_cost_bytes()
        return total

    def _device_fast_path_target(self):
        """If every member consumes onto the same CUDA device, the whole
        span goes up in ONE H2D and is sliced on the GPU — per-member H2D
        round trips dominate otherwise (measured 0.43 GB/s vs 11.3 GB/s
        for 2000 small tensors). Members opt in by implementing
        device_span_target()/consume_from_device_u8() (plain tensor and
        sharded consumers do)."""
        device = None
        for m in self.members:
            getter = getattr(m.consumer, "device_span_target", None)
            target = getter() if getter else None
            if target is None:
                return None
            if device is None:
                device = target
            elif target != device:
                return None
        return device

    def will_verify_on_device(self) -> bool:
        from .ops.staging import HIP_EXT_AVAILABLE

        return HIP_EXT_AVAILABLE and self._device_fast_path_target() is not None

    async def consume_buffer(self, ctx: StageContext, buf: BufferType) -> None:
        device = self._device_fast_path_target()
        if device is None:
            mv = memoryview(buf)
            for m in self.members:
                s, e = m.byte_range
                sub = mv[s - self.span_start : e - self.span_start]
                await m.consumer.consume_buffer(ctx, sub)
            return

        def work() -> None:
            if self._pinned_block is not None:
                # read landed in pinned memory already: straight SDMA H2D
                n = self._pinned_nbytes
                dev_span = self._pinned_block.tensor[:n].to(
                    device, non_blocking=False
                )
            else:
                mv = memoryview(buf)
                nbytes = mv.nbytes
                pool = get_pinned_pool()
                block = pool.acquire(max(nbytes, 1))
                try:
                    src = torch.frombuffer(mv, dtype=torch.uint8)
                    block.tensor[:nbytes].copy_(src)
                    dev_span = block.tensor[:nbytes].to(
                        device, non_blocking=False
                    )
                finally:
                    pool.release(block)
            if self.expected_psum is not None:
                from .ops.staging import verify_device_psum

                verify_device_psum(
                    dev_span,
                    self.expected_psum,
                    f"{self.members[0].path} span[{self.span_start}:"
                    f"{self.span_start + memoryview(buf).nbytes}]",
                )
            for m in self.members:
                s, e = m.byte_range
                sub = dev_span[s - self.span_start : e - self.span_start]
                m.consumer.consume_from_device_u8(sub)

        await asyncio.get_running_loop().run_in_executor(ctx.executor, work)
