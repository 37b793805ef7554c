"""ShardedTensor preparer + the resharding overlap machinery.

Write: each rank persists its local shards (subdivided to the max shard
size so staging/storage pipeline). Read (the resharding core): the target
tensor's local shards are intersected with the persisted shards; each
persisted shard is read once and scattered into every overlapping local
view. Restoring works at any world size / sharding spec, or into a plain
full tensor. Parity with reference torchsnapshot/io_preparers/
sharded_tensor.py:48-333.
"""

from __future__ import annotations

import asyncio
import logging
from dataclasses import dataclass
from typing import Any, List, Optional, Sequence, Tuple

import torch

from .. import knobs
from ..io_types import (
    BufferConsumer,
    BufferType,
    ReadReq,
    StageContext,
    WriteReq,
)
from ..manifest import Shard as ShardMeta
from ..manifest import ShardedTensorEntry, TensorEntry
from ..serialization import (
    dtype_to_str,
    str_to_dtype,
    tensor_from_memoryview,
    torch_load_from_bytes,
)
from .tensor import LoadFuture, TensorIOPreparer, tensor_copy

logger = logging.getLogger(__name__)


# ---------------------------------------------------------------------------
# overlap machinery (shared with the DTensor preparer)
# ---------------------------------------------------------------------------


@dataclass
class Overlap:
    """Intersection of a persisted shard and a target view, in both frames."""

    src_offsets: List[int]  # within the persisted shard
    dst_offsets: List[int]  # within the target view
    lengths: List[int]


def compute_overlap(
    src_offsets: Sequence[int],
    src_sizes: Sequence[int],
    dst_offsets: Sequence[int],
    dst_sizes: Sequence[int],
) -> Optional[Overlap]:
    so, do, ln = [], [], []
    for s_off, s_sz, d_off, d_sz in zip(
        src_offsets, src_sizes, dst_offsets, dst_sizes
    ):
        lo = max(s_off, d_off)
        hi = min(s_off + s_sz, d_off + d_sz)
        if hi <= lo:
            return None
        so.append(lo - s_off)
        do.append(lo - d_off)
        ln.append(hi - lo)
    return Overlap(src_offsets=so, dst_offsets=do, lengths=ln)


def narrow_nd(
    tensor: torch.Tensor, offsets: Sequence[int], lengths: Sequence[int]
) -> torch.Tensor:
    view = tensor
    for dim, (off, ln) in enumerate(zip(offsets, lengths)):
        view = view.narrow(dim, off, ln)
    return view


def subdivide_shard(
    tensor: torch.Tensor,
    global_offsets: Sequence[int],
    max_bytes: int,
) -> List[Tuple[torch.Tensor, List[int]]]:
    """Split a local shard along its largest dim into pieces of at most
    ``max_bytes``; returns [(piece, piece_global_offsets), ...]."""
    nbytes = tensor.numel() * tensor.element_size()
    if nbytes <= max_bytes or tensor.dim() == 0:
        return [(tensor, list(global_offsets))]
    dim = max(range(tensor.dim()), key=lambda d: tensor.shape[d])
    if tensor.shape[dim] <= 1:
        return [(tensor, list(global_offsets))]
    row_bytes = max(nbytes // tensor.shape[dim], 1)
    rows_per = max(max_bytes // row_bytes, 1)
    out: List[Tuple[torch.Tensor, List[int]]] = []
    off = 0
    while off < tensor.shape[dim]:
        ln = min(rows_per, tensor.shape[dim] - off)
        piece = tensor.narrow(dim, off, ln)
        piece_offsets = list(global_offsets)
        piece_offsets[dim] += off
        out.append((piece, piece_offsets))
        off += ln
    return out


def location_for_shard(storage_path: str, offsets: Sequence[int]) -> str:
    return f"{storage_path}.{'_'.join(str(o) for o in offsets)}"


# ---------------------------------------------------------------------------
# consumers
# ---------------------------------------------------------------------------


class ShardConsumer(BufferConsumer):
    """Deserialize one persisted shard and scatter it into every
    overlapping target view."""

    def __init__(
        self,
        shard_entry: TensorEntry,
        targets: List[Tuple[torch.Tensor, Overlap]],
    ) -> None:
        self.shard_entry = shard_entry
        self.targets = targets
        self._pinned_block = None
        self._pinned_nbytes = 0
        # (expected psum64 value, word_base) set by the read scheduler
        # when this consumer verifies on device (will_verify_on_device)
        self.expected_psum = None

    def all_targets_on_device(self) -> bool:
        return bool(self.targets) and all(
            dst.device.type == "cuda" for dst, _ in self.targets
        )

    def alloc_pinned_buffer(self, nbytes: int) -> memoryview:
        from ..ops.staging import get_pinned_pool

        self._pinned_block = get_pinned_pool().acquire(max(nbytes, 1))
        self._pinned_nbytes = nbytes
        return memoryview(self._pinned_block.tensor.numpy())[:nbytes]

    def close(self) -> None:
        if self._pinned_block is not None:
            from ..ops.staging import get_pinned_pool

            get_pinned_pool().release(self._pinned_block)
            self._pinned_block = None

    def get_consuming_cost_bytes(self) -> int:
        return self.shard_entry.nbytes_estimate()

    def device_span_target(self):
        if (
            self.all_targets_on_device()
            and self.shard_entry.serializer != "torch_save"
        ):
            return self.targets[0][0].device
        return None

    def will_verify_on_device(self) -> bool:
        from ..ops.staging import HIP_EXT_AVAILABLE

        return (
            HIP_EXT_AVAILABLE
            and self._pinned_block is not None
            and self.all_targets_on_device()
            and self.shard_entry.serializer != "torch_save"
        )

    def consume_from_device_u8(self, dev_u8: torch.Tensor) -> None:
        dtype = str_to_dtype(self.shard_entry.dtype)
        shard = (
            dev_u8.view(dtype).reshape(tuple(self.shard_entry.shape))
            if dtype != torch.uint8
            else dev_u8.reshape(tuple(self.shard_entry.shape))
        )
        for dst_view, ov in self.targets:
            src = narrow_nd(shard, ov.src_offsets, ov.lengths)
            dst = narrow_nd(dst_view, ov.dst_offsets, ov.lengths)
            tensor_copy(dst, src)

    async def consume_buffer(self, ctx: StageContext, buf: BufferType) -> None:
        def work() -> None:
            if self.shard_entry.serializer == "torch_save":
                shard = torch_load_from_bytes(bytes(buf))
            elif self._pinned_block is not None and self.all_targets_on_device():
                # storage read landed in pinned memory: straight SDMA H2D,
                # overlap scatter runs on the GPU
                n = self._pinned_nbytes
                dtype = str_to_dtype(self.shard_entry.dtype)
                dev_u8 = self._pinned_block.tensor[:n].to(
                    self.targets[0][0].device, non_blocking=False
                )
                if self.expected_psum is not None:
                    from ..ops.staging import verify_device_psum

                    verify_device_psum(
                        dev_u8, self.expected_psum, self.shard_entry.location
                    )
                shard = (
                    dev_u8.view(dtype).reshape(tuple(self.shard_entry.shape))
                    if dtype != torch.uint8
                    else dev_u8.reshape(tuple(self.shard_entry.shape))
                )
            elif self.all_targets_on_device():
                # byte-range/batched path: pinned bounce + SDMA
                from ..ops.staging import copy_buffer_via_pinned

                shard = copy_buffer_via_pinned(
                    buf,
                    dtype=str_to_dtype(self.shard_entry.dtype),
                    shape=tuple(self.shard_entry.shape),
                    device=self.targets[0][0].device,
                )
            else:
                shard = tensor_from_memoryview(
                    memoryview(buf),
                    dtype=str_to_dtype(self.shard_entry.dtype),
                    shape=tuple(self.shard_entry.shape),
                )
            for dst_view, ov in self.targets:
                src = narrow_nd(shard, ov.src_offsets, ov.lengths)
                dst = narrow_nd(dst_view, ov.dst_offsets, ov.lengths)
                tensor_copy(dst, src)

        await asyncio.get_running_loop().run_in_executor(ctx.executor, work)


def plan_shard_reads(
    shards: List[ShardMeta],
    targets: List[Tuple[torch.Tensor, Sequence[int]]],
) -> List[ReadReq]:
    """For each persisted shard overlapping any target view, emit one read.

    ``targets``: [(local_view_tensor, view_global_offsets), ...]
    """
    read_reqs: List[ReadReq] = []
    for shard in shards:
        hits: List[Tuple[torch.Tensor, Overlap]] = []
        for view, view_offsets in targets:
            ov = compute_overlap(
                shard.offsets, shard.sizes, view_offsets, list(view.shape)
            )
            if ov is not None:
                hits.append((view, ov))
        if not hits:
            continue
        byte_range = (
            tuple(shard.tensor.byte_range) if shard.tensor.byte_range else None
        )
        consumer = ShardConsumer(shard_entry=shard.tensor, targets=hits)
        buf_alloc = (
            consumer.alloc_pinned_buffer
            if (
                consumer.all_targets_on_device()
                and shard.tensor.serializer != "torch_save"
            )
            else None
        )
        read_reqs.append(
            ReadReq(
                path=shard.tensor.location,
                consumer=consumer,
                byte_range=byte_range,
                buf_alloc=buf_alloc,
            )
        )
    return read_reqs


# ---------------------------------------------------------------------------
# preparer
# ---------------------------------------------------------------------------


class ShardedTensorIOPreparer:
    @staticmethod
    def prepare_write(
        storage_path: str,
        obj: Any,  # ShardedTensor
        is_async_snapshot: bool = False,
    ) -> Tuple[ShardedTensorEntry, List[WriteReq]]:
        max_bytes = knobs.get_max_shard_size_bytes()
        shards_meta: List[ShardMeta] = []
        write_reqs: List[WriteReq] = []
        global_shape = list(obj.metadata().size)
        for local_shard in obj.local_shards():
            tensor = local_shard.tensor
            offsets = list(local_shard.metadata.shard_offsets)
            for piece, piece_offsets in subdivide_shard(
                tensor, offsets, max_bytes
            ):
                location = location_for_shard(storage_path, piece_offsets)
                sub_entry, sub_reqs = TensorIOPreparer.prepare_write(
                    storage_path=location,
                    tensor=piece,
                    replicated=False,
                    is_async_snapshot=is_async_snapshot,
                )
                shards_meta.append(
                    ShardMeta(
                        offsets=piece_offsets,
                        sizes=list(piece.shape),
                        tensor=sub_entry,
                    )
                )
                write_reqs.extend(sub_reqs)
        entry = ShardedTensorEntry(
            shards=shards_meta,
            dtype=dtype_to_str(obj.dtype),
            shape=global_shape,
        )
        return entry, write_reqs

    @staticmethod
    def prepare_read(
        entry: ShardedTensorEntry,
        obj_out: Optional[Any] = None,
    ) -> Tuple[List[ReadReq], LoadFuture]:
        try:
            from torch.distributed._shard.sharded_tensor import ShardedTensor
        except ImportError:
            ShardedTensor = ()  # type: ignore[assignment]

        global_shape = entry.shape or _infer_global_shape(entry)

        if isinstance(obj_out, ShardedTensor):
            st_shape = list(obj_out.metadata().size)
            if entry.shape and st_shape != list(entry.shape):
                logger.warning(
                    "global shape mismatch on sharded restore: snapshot %s "
                    "vs target %s; loading the overlap",
                    entry.shape,
                    st_shape,
                )
            targets = [
                (ls.tensor, list(ls.metadata.shard_offsets))
                for ls in obj_out.local_shards()
            ]
            fut = LoadFuture(obj_out)
            return plan_shard_reads(entry.shards, targets), fut

        # load into a plain full tensor (obj_out or a fresh CPU tensor)
        if isinstance(obj_out, torch.Tensor):
            full = obj_out
        else:
            dtype = (
                str_to_dtype(entry.dtype)
                if entry.dtype
                else str_to_dtype(entry.shards[0].tensor.dtype)
            )
            full = torch.empty(global_shape, dtype=dtype)
        targets = [(full, [0] * full.dim())]
        fut = LoadFuture(full)
        return plan_shard_reads(entry.shards, targets), fut


def _infer_global_shape(entry: ShardedTensorEntry) -> List[int]:
    ndim = len(entry.shards[0].offsets)
    shape = [0] * ndim
    for s in entry.shards:
        for d in range(ndim):
            shape[d] = max(shape[d], s.offsets[d] + s.sizes[d])
    return shape
