"""Entry predicates + replica-group analysis for manifest entries.

Parity with reference torchsnapshot/manifest_utils.py:36-106.
"""

from __future__ import annotations

from typing import List

import numpy as np

from .manifest import (
    DTensorEntry,
    Entry,
    is_container_entry,  # re-export for convenience
)

__all__ = [
    "is_container_entry",
    "is_sharded_entry",
    "is_replicated_entry",
    "is_fully_replicated_entry",
    "is_partially_replicated_entry",
    "get_replicated_ranks",
]


def is_sharded_entry(entry: Entry) -> bool:
    from .manifest import ShardedTensorEntry

    if isinstance(entry, ShardedTensorEntry):
        return True
    if isinstance(entry, DTensorEntry):
        return any(len(m) > 0 for m in entry.dim_map)
    return False


def is_fully_replicated_entry(entry: Entry) -> bool:
    """Replicated across ALL ranks: either marked replicated, or a DTensor
    whose every tensor dim is unsharded (all placements Replicate)."""
    if isinstance(entry, DTensorEntry):
        return all(len(m) == 0 for m in entry.dim_map)
    return bool(getattr(entry, "replicated", False))


def is_partially_replicated_entry(entry: Entry) -> bool:
    """A DTensor that is sharded along some mesh dims and replicated along
    others (e.g. HSDP)."""
    if not isinstance(entry, DTensorEntry):
        return False
    if not is_sharded_entry(entry):
        return False
    mesh = np.array(entry.mesh)
    sharded_mesh_dims = {d for m in entry.dim_map for d in m}
    return len(sharded_mesh_dims) < mesh.ndim


def is_replicated_entry(entry: Entry) -> bool:
    return is_fully_replicated_entry(entry) or is_partially_replicated_entry(entry)


def get_replicated_ranks(entry: DTensorEntry) -> List[List[int]]:
    """Group the mesh's ranks into replica sets: ranks within one set hold
    identical local shards. Sets are formed by fixing the sharded mesh dims
    and varying the replicated ones."""
    mesh = np.array(entry.mesh)
    sharded_mesh_dims = sorted({d for m in entry.dim_map for d in m})
    replicated_mesh_dims = [
        d for d in range(mesh.ndim) if d not in sharded_mesh_dims
    ]
    if not replicated_mesh_dims:
        return [[int(r)] for r in mesh.flatten()]
    # Move replicated dims last, then flatten: each row is one replica set.
    order = sharded_mesh_dims + replicated_mesh_dims
    permuted = np.transpose(mesh, order)
    n_sets = int(np.prod([mesh.shape[d] for d in sharded_mesh_dims])) if sharded_mesh_dims else 1
    rows = permuted.reshape(n_sets, -1)
    return [[int(r) for r in row] for row in rows]
