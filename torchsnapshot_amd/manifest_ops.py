"""Load-path manifest transformations.

``get_manifest_for_rank`` turns the global manifest (keys
``"<rank>/<logical_path>"``) into the view one rank uses for restore:

- this rank's own entries, prefix stripped,
- sharded entries merged across ALL writer ranks so every reader sees the
  complete shard set (DTensor replica shards deduplicated),
- replicated entries borrowed from rank 0 when this rank has none (restore
  at a larger world size than the save),

plus elasticity fix-ups for sharded tensors whose presence changed between
save and restore. Parity with reference torchsnapshot/manifest_ops.py.
"""

from __future__ import annotations

import logging
from collections import defaultdict
from typing import Dict, List, Set, Tuple

from . import knobs
from .flatten import Flattened
from .manifest import (
    DTensorEntry,
    Entry,
    Manifest,
    Shard,
    ShardedTensorEntry,
    SnapshotMetadata,
    is_container_entry,
)
from .manifest_utils import (
    is_fully_replicated_entry,
    is_sharded_entry,
)

logger = logging.getLogger(__name__)


def _split_rank_path(key: str) -> Tuple[int, str]:
    rank_str, _, path = key.partition("/")
    return int(rank_str), path


def get_manifest_for_rank(
    metadata: SnapshotMetadata, rank: int
) -> Tuple[Manifest, Dict[str, Entry]]:
    """Returns (manifest for this rank, merged flattened payload entries).

    The first mapping contains container entries + payload entries; the
    second maps logical path -> payload entry only (for read planning)."""
    world_size = metadata.world_size
    local: Manifest = {}
    # group all entries by (writing rank, logical path)
    sharded_by_path: Dict[str, List[Tuple[int, Entry]]] = defaultdict(list)
    replicated_by_path: Dict[str, List[Tuple[int, Entry]]] = defaultdict(list)
    rank0_entries: Manifest = {}

    for key, entry in metadata.manifest.items():
        w_rank, path = _split_rank_path(key)
        if is_sharded_entry(entry) or isinstance(entry, DTensorEntry):
            # every DTensor entry goes through the shard-union merge, even
            # fully-replicated ones: each replica-set rank only records the
            # shard pieces it wrote (round-robin), so no single rank's
            # entry holds the full set
            sharded_by_path[path].append((w_rank, entry))
        elif is_fully_replicated_entry(entry) and not is_container_entry(entry):
            replicated_by_path[path].append((w_rank, entry))
        if w_rank == rank:
            local[path] = entry
        if w_rank == 0:
            rank0_entries[path] = entry

    if rank >= world_size:
        # restoring at a larger world size: borrow rank 0's container
        # structure (payload entries are filled in by the replicated and
        # sharded merges below)
        for path, entry in rank0_entries.items():
            if is_container_entry(entry):
                local[path] = entry

    # replicated entries: the partitioner split writes across ranks, so the
    # authoritative entry (or entry fragments, for chunked tensors) lives
    # under the writer ranks' keys; merge them for every reader
    for path, rank_entries in replicated_by_path.items():
        rank_entries.sort(key=lambda re: re[0])
        local[path] = _merge_replicated_entries([e for _, e in rank_entries])

    # merge sharded entries so this rank sees all shards (resharding on
    # load requires every reader to know the full shard set)
    for path, rank_entries in sharded_by_path.items():
        local[path] = _merge_sharded_entries([e for _, e in rank_entries])

    return local, {p: e for p, e in local.items() if not is_container_entry(e)}


def _merge_replicated_entries(entries: List[Entry]) -> Entry:
    from .manifest import ChunkedTensorEntry

    first = entries[0]
    if isinstance(first, ChunkedTensorEntry) and len(entries) > 1:
        chunks: List[Shard] = []
        seen: Set[Tuple[int, ...]] = set()
        for e in entries:
            assert isinstance(e, ChunkedTensorEntry)
            for c in e.chunks:
                key = tuple(c.offsets)
                if key in seen:
                    continue
                seen.add(key)
                chunks.append(c)
        chunks.sort(key=lambda c: c.offsets)
        return ChunkedTensorEntry(
            dtype=first.dtype,
            shape=first.shape,
            chunks=chunks,
            replicated=True,
        )
    return first


def _merge_sharded_entries(entries: List[Entry]) -> Entry:
    first = entries[0]
    if isinstance(first, ShardedTensorEntry):
        shards: List[Shard] = []
        seen: Set[Tuple[Tuple[int, ...], Tuple[int, ...]]] = set()
        for e in entries:
            assert isinstance(e, ShardedTensorEntry)
            for s in e.shards:
                key = (tuple(s.offsets), tuple(s.sizes))
                if key in seen:
                    continue
                seen.add(key)
                shards.append(s)
        return ShardedTensorEntry(
            shards=shards, dtype=first.dtype, shape=first.shape
        )
    if isinstance(first, DTensorEntry):
        # DTensor with partial replication: multiple ranks hold identical
        # shards; keep one copy of each (offset, size) region.
        shards = []
        seen = set()
        for e in entries:
            assert isinstance(e, DTensorEntry)
            for s in e.shards:
                key = (tuple(s.offsets), tuple(s.sizes))
                if key in seen:
                    continue
                seen.add(key)
                shards.append(s)
        return DTensorEntry(
            shards=shards,
            mesh=first.mesh,
            dim_map=first.dim_map,
            dtype=first.dtype,
            shape=first.shape,
        )
    raise TypeError(f"cannot merge entries of type {type(first).__name__}")


def remove_entry_from_manifest(manifest: Manifest, path: str) -> None:
    """Remove an entry AND unlink it from its parent container entry so the
    manifest stays inflatable (parity with reference
    torchsnapshot/manifest_ops.py:250-288)."""
    from .flatten import unescape_key

    manifest.pop(path, None)
    if "/" not in path:
        return
    parent_path, _, segment = path.rpartition("/")
    parent = manifest.get(parent_path)
    if parent is None:
        return
    from .manifest import DictEntry, ListEntry

    if isinstance(parent, DictEntry):  # covers OrderedDictEntry
        key = unescape_key(segment)
        for cand in (key, int(key) if key.lstrip("-").isdigit() else key):
            if cand in parent.keys:
                parent.keys.remove(cand)
                break
    elif isinstance(parent, ListEntry):
        # a list with a removed element cannot be reindexed faithfully;
        # degrade the parent to a dict-like view is not possible either —
        # refuse, matching the reference's container constraints
        raise ValueError(
            f"cannot remove '{path}': parent '{parent_path}' is a list"
        )


def handle_sharded_tensor_elasticity(
    rank_manifest: Manifest,
    payload_entries: Dict[str, Entry],
    flattened_target: Flattened,
) -> None:
    """Reconcile sharded-tensor presence differences between the snapshot
    and the restore-time state dict (world-size changes can make per-rank
    optimizer states appear/disappear).

    - a sharded entry in the snapshot with no matching target path is
      dropped (the target doesn't want it),
    - a target path that is a ShardedTensor but has no snapshot entry is
      dropped from the load set with a warning (it keeps its init values),
      rather than failing the whole restore.

    Mirrors reference torchsnapshot/manifest_ops.py:180-247 in effect.
    """
    from .dtensor_utils import is_sharded as _runtime_is_sharded

    root_only = knobs.is_sharded_elasticity_root_only()

    for path in list(payload_entries.keys()):
        entry = payload_entries[path]
        if not is_sharded_entry(entry):
            continue
        if path not in flattened_target:
            if root_only and "/" in path:
                continue
            logger.info(
                "sharded entry '%s' not requested by the target state dict; "
                "skipping",
                path,
            )
            del payload_entries[path]
            remove_entry_from_manifest(rank_manifest, path)

    for path, obj in list(flattened_target.items()):
        if _runtime_is_sharded(obj) and path not in payload_entries:
            logger.warning(
                "target sharded tensor '%s' has no entry in the snapshot; "
                "it will keep its current values",
                path,
            )
            del flattened_target[path]


def get_available_entries(metadata: SnapshotMetadata, rank: int) -> Manifest:
    """The manifest view ``read_object``/``get_manifest`` expose for a rank."""
    manifest, _ = get_manifest_for_rank(metadata, rank)
    return manifest
