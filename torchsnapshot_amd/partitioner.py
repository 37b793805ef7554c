"""Write-load distribution for replicated objects.

Every rank holds an identical copy of a replicated object, so exactly one
rank should write it. Rank 0 greedily assigns each replicated write request
(chunked tensors are assigned chunk-by-chunk, so one 20 GB replicated
model spreads across all ranks) to the currently least-loaded rank, seeded
with each rank's unavoidable non-replicated write load; the assignment is
broadcast so all ranks agree.

After partitioning, a rank's manifest keeps a replicated entry only for
payloads it actually writes (a partial ChunkedTensorEntry when it writes a
subset of chunks); on load, manifest_ops re-merges the partial entries.
This replaces the reference's consolidate_replicated_entries collective
(torchsnapshot/partitioner.py:216-355) with a merge on the read side.
"""

from __future__ import annotations

import logging
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence

from .pg_wrapper import PGWrapper

logger = logging.getLogger(__name__)


@dataclass
class PartitionItem:
    """One assignable replicated write request."""

    req_path: str
    nbytes: int
    # ranks allowed to write this payload (None = any rank); used for
    # partially-replicated DTensor shards, which exist only on their
    # replica set
    allowed_ranks: Optional[List[int]] = None


def partition_write_reqs(
    items: Sequence[PartitionItem],
    my_non_replicated_bytes: int,
    pg: PGWrapper,
) -> Dict[str, int]:
    """Returns {req_path: writer_rank} for every replicated request."""
    world_size = pg.get_world_size()
    if world_size == 1:
        return {item.req_path: 0 for item in items}

    loads: List[Optional[int]] = [None] * world_size
    pg.all_gather_object(loads, my_non_replicated_bytes)

    payload: List[Optional[Dict[str, int]]] = [None]
    if pg.get_rank() == 0:
        payload[0] = _greedy_assign(items, [int(l or 0) for l in loads])
    pg.broadcast_object_list(payload, src=0)
    assignment = payload[0]
    assert assignment is not None
    return assignment


def _greedy_assign(
    items: Sequence[PartitionItem], loads: List[int]
) -> Dict[str, int]:
    assignment: Dict[str, int] = {}
    # big payloads first so they land on distinct ranks
    for item in sorted(items, key=lambda it: it.nbytes, reverse=True):
        candidates = (
            item.allowed_ranks
            if item.allowed_ranks is not None
            else range(len(loads))
        )
        writer = min(candidates, key=lambda r: loads[r])
        assignment[item.req_path] = writer
        loads[writer] += item.nbytes
    return assignment
