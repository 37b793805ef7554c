"""Chunked-tensor preparer: large tensors split along dim 0 so staging and
storage I/O pipeline chunk-by-chunk instead of stalling on one huge buffer.

Parity with reference torchsnapshot/io_preparers/chunked_tensor.py:36-128.
Chunk payloads live at ``<storage_path>_c<offset>``.
"""

from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch

from .. import knobs
from ..io_types import ReadReq, WriteReq
from ..manifest import ChunkedTensorEntry, Shard
from ..serialization import dtype_to_str, str_to_dtype
from .tensor import LoadFuture, TensorIOPreparer, tensor_copy


def should_chunk(tensor: torch.Tensor) -> bool:
    nbytes = tensor.numel() * tensor.element_size()
    return (
        not tensor.is_quantized
        and tensor.dim() > 0
        and tensor.shape[0] > 1
        and nbytes > knobs.get_max_chunk_size_bytes()
    )


def _chunk_plan(tensor_shape: List[int], elem_size: int) -> List[Tuple[int, int]]:
    """[(offset, length), ...] along dim 0 so each chunk is at most the max
    chunk size (last chunk may be smaller)."""
    dim0 = tensor_shape[0]
    row_bytes = elem_size * math.prod(tensor_shape[1:]) if len(tensor_shape) > 1 else elem_size
    max_bytes = knobs.get_max_chunk_size_bytes()
    rows_per_chunk = max(max_bytes // max(row_bytes, 1), 1)
    plan = []
    off = 0
    while off < dim0:
        length = min(rows_per_chunk, dim0 - off)
        plan.append((off, length))
        off += length
    return plan


class ChunkedTensorIOPreparer:
    @staticmethod
    def prepare_write(
        storage_path: str,
        tensor: torch.Tensor,
        replicated: bool = False,
        is_async_snapshot: bool = False,
    ) -> Tuple[ChunkedTensorEntry, List[WriteReq]]:
        plan = _chunk_plan(list(tensor.shape), tensor.element_size())
        chunks: List[Shard] = []
        write_reqs: List[WriteReq] = []
        ndim = tensor.dim()
        for off, length in plan:
            chunk = tensor.narrow(0, off, length)
            location = f"{storage_path}_c{off}"
            sub_entry, sub_reqs = TensorIOPreparer.prepare_write(
                storage_path=location,
                tensor=chunk,
                replicated=replicated,
                is_async_snapshot=is_async_snapshot,
            )
            chunks.append(
                Shard(
                    offsets=[off] + [0] * (ndim - 1),
                    sizes=list(chunk.shape),
                    tensor=sub_entry,
                )
            )
            write_reqs.extend(sub_reqs)
        entry = ChunkedTensorEntry(
            dtype=dtype_to_str(tensor.dtype),
            shape=list(tensor.shape),
            chunks=chunks,
            replicated=replicated,
        )
        return entry, write_reqs

    @staticmethod
    def prepare_read(
        entry: ChunkedTensorEntry,
        tensor_out: Optional[torch.Tensor] = None,
        buffer_size_limit_bytes: Optional[int] = None,
    ) -> Tuple[List[ReadReq], LoadFuture]:
        if tensor_out is None or list(tensor_out.shape) != list(entry.shape):
            # shape mismatch (elasticity) or no target: stage into a fresh
            # tensor, then copy what overlaps at the end if a target exists
            staging = torch.empty(
                entry.shape, dtype=str_to_dtype(entry.dtype)
            )
            dst = staging
            final_out = tensor_out
        else:
            dst = tensor_out
            staging = None
            final_out = None
        fut = LoadFuture(tensor_out if tensor_out is not None else dst)
        read_reqs: List[ReadReq] = []
        chunk_futs: List[LoadFuture] = []
        for chunk in entry.chunks:
            view = dst.narrow(0, chunk.offsets[0], chunk.sizes[0])
            sub_reqs, sub_fut = TensorIOPreparer.prepare_read(
                chunk.tensor, view, buffer_size_limit_bytes
            )
            read_reqs.extend(sub_reqs)
            chunk_futs.append(sub_fut)
        if final_out is not None and read_reqs:
            # wrap the final request's consumer to trigger the last copy
            _chain_final_copy(read_reqs, staging, final_out)
        return read_reqs, fut


def _chain_final_copy(
    read_reqs: List[ReadReq], staging: torch.Tensor, final_out: torch.Tensor
) -> None:
    import threading

    lock = threading.Lock()
    remaining = [len(read_reqs)]

    for rr in read_reqs:
        orig = rr.consumer.consume_buffer

        async def wrapped(ctx, buf, _orig=orig):
            await _orig(ctx, buf)
            with lock:
                remaining[0] -= 1
                last = remaining[0] == 0
            if last:
                tensor_copy(final_out, staging)

        rr.consumer.consume_buffer = wrapped  # type: ignore[method-assign]
