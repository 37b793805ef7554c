"""Dispatch: runtime object -> (manifest entry, write reqs) and
manifest entry -> (read reqs, future).

Parity with reference torchsnapshot/io_preparer.py:52-182. Storage-path
categories determine the payload file prefix:

- per-rank objects:            ``<rank>/<logical_path>``
- replicated objects:          ``replicated/<logical_path>``
- sharded tensors:             ``sharded/<logical_path>``
- partially-replicated DTensor:``replicated_sharded/<logical_path>``
"""

from __future__ import annotations

from typing import Any, List, Optional, Tuple

import torch

from .io_types import ReadReq, WriteReq
from .manifest import (
    ChunkedTensorEntry,
    DTensorEntry,
    Entry,
    ObjectEntry,
    PrimitiveEntry,
    ShardedTensorEntry,
    TensorEntry,
)
from .io_preparers.chunked_tensor import ChunkedTensorIOPreparer, should_chunk
from .io_preparers.object import ObjectIOPreparer
from .io_preparers.tensor import LoadFuture, TensorIOPreparer


def _is_sharded_tensor(obj: Any) -> bool:
    try:
        from torch.distributed._shard.sharded_tensor import ShardedTensor

        return isinstance(obj, ShardedTensor)
    except ImportError:
        return False


def _is_dtensor(obj: Any) -> bool:
    try:
        from torch.distributed.tensor import DTensor

        return isinstance(obj, DTensor)
    except ImportError:
        return False


def get_storage_path(
    obj: Any, logical_path: str, rank: int, replicated: bool
) -> str:
    if _is_sharded_tensor(obj):
        return f"sharded/{logical_path}"
    if _is_dtensor(obj):
        from .dtensor_utils import is_sharded_dtensor

        if is_sharded_dtensor(obj):
            return f"replicated_sharded/{logical_path}"
        return f"replicated/{logical_path}"
    if replicated:
        return f"replicated/{logical_path}"
    return f"{rank}/{logical_path}"


def prepare_write(
    obj: Any,
    logical_path: str,
    rank: int,
    replicated: bool = False,
    is_async_snapshot: bool = False,
) -> Tuple[Entry, List[WriteReq]]:
    """Build the manifest entry + write requests for one flattened leaf.

    No data moves here — stagers capture references; the scheduler drives
    the actual staging and storage I/O later."""
    if PrimitiveEntry.supported(obj):
        return PrimitiveEntry.from_object(obj, replicated), []

    storage_path = get_storage_path(obj, logical_path, rank, replicated)

    if _is_sharded_tensor(obj):
        from .io_preparers.sharded_tensor import ShardedTensorIOPreparer

        return ShardedTensorIOPreparer.prepare_write(
            storage_path, obj, is_async_snapshot=is_async_snapshot
        )
    if _is_dtensor(obj):
        from .io_preparers.dtensor import DTensorIOPreparer

        return DTensorIOPreparer.prepare_write(
            storage_path, obj, is_async_snapshot=is_async_snapshot
        )
    if isinstance(obj, torch.Tensor):
        if should_chunk(obj):
            return ChunkedTensorIOPreparer.prepare_write(
                storage_path,
                obj,
                replicated=replicated,
                is_async_snapshot=is_async_snapshot,
            )
        return TensorIOPreparer.prepare_write(
            storage_path,
            obj,
            replicated=replicated,
            is_async_snapshot=is_async_snapshot,
        )
    return ObjectIOPreparer.prepare_write(storage_path, obj, replicated=replicated)


def prepare_read(
    entry: Entry,
    obj_out: Optional[Any] = None,
    buffer_size_limit_bytes: Optional[int] = None,
) -> Tuple[List[ReadReq], LoadFuture]:
    """Build read requests that load ``entry`` (into ``obj_out`` in-place
    when possible)."""
    if isinstance(entry, PrimitiveEntry):
        return [], LoadFuture(entry.get_value())
    if isinstance(entry, ShardedTensorEntry):
        from .io_preparers.sharded_tensor import ShardedTensorIOPreparer

        return ShardedTensorIOPreparer.prepare_read(entry, obj_out)
    if isinstance(entry, DTensorEntry):
        from .io_preparers.dtensor import DTensorIOPreparer

        return DTensorIOPreparer.prepare_read(entry, obj_out)
    if isinstance(entry, ChunkedTensorEntry):
        tensor_out = obj_out if isinstance(obj_out, torch.Tensor) else None
        return ChunkedTensorIOPreparer.prepare_read(
            entry, tensor_out, buffer_size_limit_bytes
        )
    if isinstance(entry, TensorEntry):
        tensor_out = obj_out if isinstance(obj_out, torch.Tensor) else None
        return TensorIOPreparer.prepare_read(
            entry, tensor_out, buffer_size_limit_bytes
        )
    if isinstance(entry, ObjectEntry):
        return ObjectIOPreparer.prepare_read(entry)
    raise TypeError(f"cannot prepare read for entry type {type(entry).__name__}")
