"""DTensor preparer: sharded + (partially) replicated tensors on device
meshes (FSDP2 / HSDP / TP / SP layouts).

Write: each rank computes its local shard's global coordinates; within a
replica set (ranks holding identical shards) exactly one member writes each
payload, chosen round-robin by sub-shard index so replicated write load
spreads across the set (the reference routes this through its partitioner,
torchsnapshot/partitioner.py:90-104; here the choice is deterministic from
the mesh, saving a collective). All replicas emit identical entries; the
load-side merge dedups them by offsets (manifest_ops).

Read: the same overlap machinery as ShardedTensor, against
``obj_out.to_local()``. Parity with reference torchsnapshot/io_preparers/
dtensor.py:63-278.
"""

from __future__ import annotations

import logging
from typing import Any, List, Optional, Tuple

import torch

from .. import knobs
from ..io_types import ReadReq, WriteReq
from ..manifest import DTensorEntry, Shard as ShardMeta
from ..serialization import dtype_to_str, str_to_dtype
from .sharded_tensor import (
    location_for_shard,
    plan_shard_reads,
    subdivide_shard,
)
from .tensor import LoadFuture, TensorIOPreparer

logger = logging.getLogger(__name__)


def _dim_map(dt: Any) -> List[List[int]]:
    from torch.distributed.tensor.placement_types import Shard

    dim_map: List[List[int]] = [[] for _ in range(dt.ndim)]
    for mesh_dim, placement in enumerate(dt.placements):
        if isinstance(placement, Shard):
            dim_map[placement.dim].append(mesh_dim)
    return dim_map


def _local_global_offset(dt: Any) -> Tuple[List[int], List[int]]:
    """(local shape, global offset) of this rank's shard of the DTensor."""
    from torch.distributed.tensor._utils import (
        compute_local_shape_and_global_offset,
    )

    local_shape, global_offset = compute_local_shape_and_global_offset(
        dt.shape, dt.device_mesh, dt.placements
    )
    return list(local_shape), list(global_offset)


_replica_set_cache: dict = {}


def _my_replica_set(dt: Any) -> List[int]:
    """Global ranks that hold a shard identical to this rank's (the ranks
    reached by varying only replicated mesh dims at this rank's mesh
    coordinate). Cached per (mesh, placements): a model checkpoint calls
    this once per parameter with identical inputs, and the mesh tensor
    math is measurable in the async-stall window under CPU contention."""
    from torch.distributed.tensor.placement_types import Shard

    mesh = dt.device_mesh.mesh
    cache_key = (
        tuple(mesh.flatten().tolist()),
        tuple(mesh.shape),
        tuple(str(p) for p in dt.placements),
        torch.distributed.get_rank() if torch.distributed.is_initialized() else 0,
    )
    hit = _replica_set_cache.get(cache_key)
    if hit is not None:
        return hit
    result = _my_replica_set_impl(dt)
    if len(_replica_set_cache) < 64:
        _replica_set_cache[cache_key] = result
    return result


def _my_replica_set_impl(dt: Any) -> List[int]:
    from torch.distributed.tensor.placement_types import Shard

    mesh = dt.device_mesh.mesh
    sharded_mesh_dims = {
        mesh_dim
        for mesh_dim, p in enumerate(dt.placements)
        if isinstance(p, Shard)
    }
    rank = torch.distributed.get_rank() if torch.distributed.is_initialized() else 0
    coord = (mesh == rank).nonzero()
    if coord.numel() == 0:
        return [rank]
    coord = coord[0].tolist()
    # vary replicated dims, fix sharded dims
    index: List[Any] = []
    for d in range(mesh.dim()):
        if d in sharded_mesh_dims:
            index.append(coord[d])
        else:
            index.append(slice(None))
    subset = mesh[tuple(index)]
    return [int(r) for r in subset.flatten().tolist()]


class DTensorIOPreparer:
    @staticmethod
    def prepare_write(
        storage_path: str,
        obj: Any,  # DTensor
        is_async_snapshot: bool = False,
    ) -> Tuple[DTensorEntry, List[WriteReq]]:
        local = obj.to_local()
        local_shape, global_offset = _local_global_offset(obj)
        if list(local.shape) != local_shape:
            # padded uneven shard (torch pads the last rank): trim
            local = local[tuple(slice(0, s) for s in local_shape)]

        replica_set = sorted(_my_replica_set(obj))
        my_rank = (
            torch.distributed.get_rank()
            if torch.distributed.is_initialized()
            else 0
        )
        max_bytes = knobs.get_max_shard_size_bytes()

        shards_meta: List[ShardMeta] = []
        write_reqs: List[WriteReq] = []
        pieces = subdivide_shard(local, global_offset, max_bytes)
        for i, (piece, piece_offsets) in enumerate(pieces):
            # round-robin writer within the replica set. ONLY the writer
            # emits the shard entry: non-writer copies would keep the
            # standalone location after the writer's batcher relocates the
            # payload into a slab, and the load-side merge could pick the
            # stale copy (round-1 advisor finding). The load-side merge
            # unions shards across all ranks, so readers still see the
            # full shard set.
            writer = replica_set[i % len(replica_set)]
            if writer != my_rank:
                continue
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
        entry = DTensorEntry(
            shards=shards_meta,
            mesh=obj.device_mesh.mesh.tolist(),
            dim_map=_dim_map(obj),
            dtype=dtype_to_str(obj.dtype),
            shape=list(obj.shape),
        )
        return entry, write_reqs

    @staticmethod
    def prepare_read(
        entry: DTensorEntry,
        obj_out: Optional[Any] = None,
    ) -> Tuple[List[ReadReq], LoadFuture]:
        try:
            from torch.distributed.tensor import DTensor
        except ImportError:
            DTensor = ()  # type: ignore[assignment]

        if isinstance(obj_out, DTensor):
            local = obj_out.to_local()
            local_shape, global_offset = _local_global_offset(obj_out)
            if list(local.shape) != local_shape:
                local = local[tuple(slice(0, s) for s in local_shape)]
            targets = [(local, global_offset)]
            fut = LoadFuture(obj_out)
            return plan_shard_reads(entry.shards, targets), fut

        if isinstance(obj_out, torch.Tensor):
            full = obj_out
        else:
            full = torch.empty(entry.shape, dtype=str_to_dtype(entry.dtype))
        targets = [(full, [0] * full.dim())]
        fut = LoadFuture(full)
        return plan_shard_reads(entry.shards, targets), fut
