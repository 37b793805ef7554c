"""Snapshot: the user-facing API.

    Snapshot.take(path, app_state)          — synchronous save
    Snapshot.async_take(path, app_state)    — non-blocking save (training
                                              resumes once staging is done)
    Snapshot(path).restore(app_state)       — in-place load
    Snapshot(path).read_object(path)        — random access to one object
    Snapshot(path).get_manifest()           — inspection

Commit protocol: all ranks write payloads, then the ``.snapshot_metadata``
manifest is written by rank 0 only after a barrier — a snapshot without
metadata is incomplete by construction (atomicity; parity with reference
torchsnapshot/snapshot.py:202-209). Async snapshots synchronize the commit
through a store-based LinearBarrier because collectives must not run on
the background thread (reference snapshot.py:1010-1032).

Collectives used (all small object collectives over RCCL/gloo): path
broadcast, replicated-glob intersection, key-set union, partitioner
assignment, global-manifest gather. Payload bytes never cross ranks.
"""

from __future__ import annotations

import fnmatch
import logging
import threading
import uuid
from dataclasses import replace as dataclass_replace
from typing import Any, Dict, List, Optional, Set, Tuple

import torch
import torch.distributed as dist

from . import knobs
from .dist_store import LinearBarrier, get_or_create_store
from .event import Event, log_event
from .flatten import Flattened, flatten, inflate
from .io_preparer import prepare_read, prepare_write
from .io_types import ReadReq, StoragePlugin, WriteIO, WriteReq
from .manifest import (
    ChunkedTensorEntry,
    Entry,
    Manifest,
    METADATA_FILENAME,
    PrimitiveEntry,
    SnapshotMetadata,
)
from .manifest_ops import (
    get_manifest_for_rank,
    handle_sharded_tensor_elasticity,
)
from .partitioner import PartitionItem, partition_write_reqs
from .pg_wrapper import PGWrapper
from .rng_state import RNGState
from .scheduler import (
    PendingIOWork,
    execute_write_reqs,
    get_process_memory_budget_bytes,
    sync_execute_read_reqs,
)
from .stateful import AppState, Stateful
from .storage import url_to_storage_plugin
from .version import __version__

logger = logging.getLogger(__name__)


# ---------------------------------------------------------------------------
# glob matching ("**" crosses path segments, "*" does not)
# ---------------------------------------------------------------------------


def glob_match(path: str, pattern: str) -> bool:
    return _match_segments(path.split("/"), pattern.split("/"))


def _match_segments(segs: List[str], pats: List[str]) -> bool:
    if not pats:
        return not segs
    head, rest = pats[0], pats[1:]
    if head == "**":
        # try consuming 0..len(segs) segments
        for i in range(len(segs) + 1):
            if _match_segments(segs[i:], rest):
                return True
        return False
    if not segs:
        return False
    if not fnmatch.fnmatchcase(segs[0], head):
        return False
    return _match_segments(segs[1:], rest)


# ---------------------------------------------------------------------------
# Snapshot
# ---------------------------------------------------------------------------


class Snapshot:
    def __init__(
        self,
        path: str,
        pg: Optional[dist.ProcessGroup] = None,
        storage_options: Optional[Dict[str, Any]] = None,
    ) -> None:
        self.path = path
        self.pg = pg
        self._storage_options = storage_options
        self._metadata: Optional[SnapshotMetadata] = None

    # -- inspection ---------------------------------------------------------

    @property
    def metadata(self) -> SnapshotMetadata:
        if self._metadata is None:
            storage = url_to_storage_plugin(self.path, self._storage_options)
            try:
                from .io_types import ReadIO

                read_io = ReadIO(path=METADATA_FILENAME)
                try:
                    storage.sync_read(read_io)
                except FileNotFoundError:
                    raise RuntimeError(
                        f"{self.path} is not a valid snapshot: it is either "
                        "incomplete or corrupted (missing "
                        f"{METADATA_FILENAME})"
                    ) from None
                self._metadata = SnapshotMetadata.from_str(
                    bytes(read_io.buf).decode("utf-8")
                )
                from .version import __version__ as _v

                if (
                    self._metadata.version.split(".")[0]
                    != _v.split(".")[0]
                ):
                    logger.warning(
                        "snapshot at %s was written by version %s; this "
                        "build is %s — load will be attempted anyway",
                        self.path,
                        self._metadata.version,
                        _v,
                    )
            finally:
                storage.sync_close()
        return self._metadata

    def get_manifest(self) -> Dict[str, Any]:
        """A deep copy of the global manifest as plain dicts."""
        return {k: v.to_dict() for k, v in self.metadata.manifest.items()}

    def delete(self) -> None:
        """Delete the snapshot (metadata first, so a concurrent reader sees
        an invalid snapshot rather than a partially-deleted one)."""
        storage = url_to_storage_plugin(self.path, self._storage_options)
        try:
            from .scheduler import run_coro_sync

            async def go():
                try:
                    await storage.delete(METADATA_FILENAME)
                except FileNotFoundError:
                    pass
                await storage.delete_dir("")

            run_coro_sync(go())
        finally:
            storage.sync_close()
        self._metadata = None

    # -- save ---------------------------------------------------------------

    @classmethod
    def take(
        cls,
        path: str,
        app_state: AppState,
        pg: Optional[dist.ProcessGroup] = None,
        replicated: Optional[List[str]] = None,
        storage_options: Optional[Dict[str, Any]] = None,
        _custom_tensor_prepare_func: Optional[Any] = None,
    ) -> "Snapshot":
        torch._C._log_api_usage_once("torchsnapshot_amd.Snapshot.take")
        cls._validate_app_state(app_state)
        pg_wrapper = PGWrapper(pg)
        unique_id = uuid.uuid4().hex
        event_meta = {"id": unique_id, "rank": pg_wrapper.get_rank(), "api": "take"}
        log_event(Event("take_start", dict(event_meta)))
        try:
            path, replicated, _ = cls._coalesce_path_and_replicated(
                path, app_state, replicated or [], pg_wrapper
            )
            storage = url_to_storage_plugin(path, storage_options)
            try:
                pending_io_work, metadata = cls._take_impl(
                    path=path,
                    app_state=app_state,
                    storage=storage,
                    pg_wrapper=pg_wrapper,
                    replicated=replicated,
                    is_async=False,
                    custom_tensor_prepare_func=_custom_tensor_prepare_func,
                )
                pending_io_work.complete()
                from .integrity import write_checksum_file

                write_checksum_file(
                    storage, pg_wrapper.get_rank(), pending_io_work.checksums
                )
                cls._commit(storage, metadata, pg_wrapper)
            finally:
                storage.sync_close()
            snapshot = cls(path, pg, storage_options)
            snapshot._metadata = metadata
            log_event(Event("take_end", {**event_meta, "success": True}))
            return snapshot
        except Exception:
            log_event(Event("take_end", {**event_meta, "success": False}))
            raise

    @classmethod
    def async_take(
        cls,
        path: str,
        app_state: AppState,
        pg: Optional[dist.ProcessGroup] = None,
        replicated: Optional[List[str]] = None,
        storage_options: Optional[Dict[str, Any]] = None,
        _custom_tensor_prepare_func: Optional[Any] = None,
    ) -> "PendingSnapshot":
        torch._C._log_api_usage_once("torchsnapshot_amd.Snapshot.async_take")
        cls._validate_app_state(app_state)
        pg_wrapper = PGWrapper(pg)
        unique_id = uuid.uuid4().hex
        event_meta = {
            "id": unique_id,
            "rank": pg_wrapper.get_rank(),
            "api": "async_take",
        }
        log_event(Event("async_take_start", dict(event_meta)))
        path, replicated, snapshot_uid = cls._coalesce_path_and_replicated(
            path, app_state, replicated or [], pg_wrapper
        )
        storage = url_to_storage_plugin(path, storage_options)
        # the commit barrier store must be created on the main thread
        # (bootstrap may use a collective); its key namespace is unique per
        # snapshot so repeated saves to the same path (or a save after a
        # failed one) never see stale counters or a stale error_flag
        store = get_or_create_store(pg_wrapper)
        barrier = LinearBarrier(
            prefix=f"tsamd_commit_{snapshot_uid}",
            store=store,
            rank=pg_wrapper.get_rank(),
            world_size=pg_wrapper.get_world_size(),
        )
        try:
            pending_io_work, metadata = cls._take_impl(
                path=path,
                app_state=app_state,
                storage=storage,
                pg_wrapper=pg_wrapper,
                replicated=replicated,
                is_async=True,
                custom_tensor_prepare_func=_custom_tensor_prepare_func,
            )
            # once staging is complete, the app may mutate its state
            # freely. With shadow clones (sources_immutable) there is
            # nothing to wait for — the pipeline owns private copies.
            if not pending_io_work.sources_immutable:
                pending_io_work.wait_staged()
        except Exception as e:
            # peers are (or will be) waiting in the commit barrier: tell
            # them this rank failed before they time out
            try:
                barrier.report_error(e)
            except Exception:
                logger.exception("failed to report early async-take error")
            log_event(
                Event("async_take_end", {**event_meta, "success": False})
            )
            raise
        return PendingSnapshot(
            path=path,
            pending_io_work=pending_io_work,
            pg_wrapper=pg_wrapper,
            metadata=metadata,
            storage=storage,
            barrier=barrier,
            storage_options=storage_options,
            event_meta=event_meta,
        )

    @classmethod
    def _take_impl(
        cls,
        path: str,
        app_state: AppState,
        storage: StoragePlugin,
        pg_wrapper: PGWrapper,
        replicated: List[str],
        is_async: bool,
        custom_tensor_prepare_func: Optional[Any] = None,
    ) -> Tuple[PendingIOWork, SnapshotMetadata]:
        rank = pg_wrapper.get_rank()
        world_size = pg_wrapper.get_world_size()

        # RNG invariance: capture the RNG state before any state_dict()
        # call can perturb it; restore it afterwards.
        rng_state_captured = torch.get_rng_state()
        has_rng_stateful = any(
            isinstance(v, RNGState) for v in app_state.values()
        )

        all_keys = cls._gather_keys(app_state, pg_wrapper)
        manifest: Manifest = {}
        flattened: Flattened = {}
        for key in all_keys:
            if key in app_state:
                sd = app_state[key].state_dict()
                m, f = flatten(sd, prefix=key)
                manifest.update(m)
                flattened.update(f)
            # collectives inside state_dict() must not interleave across
            # statefuls on different ranks
            pg_wrapper.barrier()

        if has_rng_stateful:
            torch.set_rng_state(rng_state_captured)

        replicated_paths = cls._calculate_replicated_entries(
            flattened, replicated, pg_wrapper
        )

        if custom_tensor_prepare_func is not None:
            # save-time tensor transform hook (e.g. cast fp32 weights to
            # bf16 in the snapshot); mirrors the reference's private
            # _custom_tensor_prepare_func (snapshot.py take/_take_impl)
            for p in list(flattened.keys()):
                obj = flattened[p]
                if isinstance(obj, torch.Tensor):
                    flattened[p] = custom_tensor_prepare_func(p, obj)

        # zero-stall async saves: shadow-clone the state before staging so
        # training may resume immediately (device clones run at HBM rate —
        # milliseconds for a whole model on 288 GB HBM3E; vs waiting for
        # the full D2H staging pass). Cloning preserves aliasing, so
        # tied-weight dedup below still sees shared storage.
        sources_immutable = False
        if is_async:
            sources_immutable = cls._shadow_for_async(flattened)

        write_reqs: List[WriteReq] = []
        req_to_logical: Dict[str, str] = {}
        # tied-weight dedup: identical tensor objects (same storage, view,
        # dtype — e.g. tied input/output embeddings) are written once; the
        # aliases SHARE the writer's entry object, so batcher relocation
        # stays consistent for every path
        seen_tensors: Dict[Any, Entry] = {}
        for logical_path, obj in flattened.items():
            alias_key = None
            if (
                isinstance(obj, torch.Tensor)
                and type(obj) is torch.Tensor
                and not obj.is_quantized
                and obj.layout == torch.strided  # sparse has no data_ptr
            ):
                alias_key = (
                    obj.data_ptr(),
                    obj.dtype,
                    tuple(obj.shape),
                    tuple(obj.stride()),
                    str(obj.device),
                    logical_path in replicated_paths,
                )
                if alias_key in seen_tensors:
                    manifest[logical_path] = seen_tensors[alias_key]
                    continue
            entry, reqs = prepare_write(
                obj=obj,
                logical_path=logical_path,
                rank=rank,
                replicated=logical_path in replicated_paths,
                is_async_snapshot=is_async,
            )
            manifest[logical_path] = entry
            if alias_key is not None:
                seen_tensors[alias_key] = entry
            for r in reqs:
                req_to_logical[r.path] = logical_path
            write_reqs.extend(reqs)

        # replicated write-load balancing
        if world_size > 1 and not knobs.is_partitioner_disabled():
            write_reqs = cls._partition_replicated(
                manifest, write_reqs, req_to_logical, replicated_paths,
                pg_wrapper,
            )

        write_reqs = _batch(write_reqs, rank)

        global_manifest = cls._gather_manifest(manifest, pg_wrapper)
        metadata = SnapshotMetadata(
            version=__version__,
            world_size=world_size,
            manifest=global_manifest,
        )

        budget = get_process_memory_budget_bytes(pg_wrapper)
        pending = execute_write_reqs(
            write_reqs, storage, budget, rank=rank, is_async=is_async
        )
        pending.sources_immutable = sources_immutable
        return pending, metadata

    @classmethod
    def _shadow_for_async(cls, flattened: Flattened) -> bool:
        """Replace every tensor leaf with a private clone (device clones
        at HBM rate; identical tensors share ONE clone so tied-weight
        dedup survives). Returns True iff EVERY leaf is now immutable-safe
        — then async_take may return without waiting for staging. Falls
        back (False) when a leaf can't be shadowed (ShardedTensor,
        arbitrary objects), device memory is too tight, or host-tensor
        bytes exceed the eager-clone budget."""
        mode = knobs.get_async_shadow_mode()
        if mode == "0":
            return False
        try:
            from torch.distributed.tensor import DTensor
        except ImportError:
            DTensor = ()  # type: ignore[assignment]

        tensor_paths: List[str] = []
        cuda_need: Dict[int, int] = {}
        cpu_bytes = 0
        try:
            return cls._classify_and_shadow(
                flattened, tensor_paths, cuda_need, cpu_bytes, mode, DTensor
            )
        except Exception:
            # exotic leaves (sparse tensors raise on data_ptr, etc.):
            # fall back to the classic wait-for-staging path
            logger.debug("async shadow disabled by exception", exc_info=True)
            return False

    @classmethod
    def _classify_and_shadow(
        cls,
        flattened: Flattened,
        tensor_paths: List[str],
        cuda_need: Dict[int, int],
        cpu_bytes: int,
        mode: str,
        DTensor: Any,
    ) -> bool:
        for p, obj in flattened.items():
            if isinstance(obj, (int, float, str, bool, bytes)) or obj is None:
                continue
            if DTensor and isinstance(obj, DTensor):
                local = obj.to_local()
                nbytes = local.numel() * local.element_size()
                if local.device.type == "cuda":
                    idx = local.device.index or 0
                    cuda_need[idx] = cuda_need.get(idx, 0) + nbytes
                else:
                    cpu_bytes += nbytes
                tensor_paths.append(p)
            elif isinstance(obj, torch.Tensor) and (
                type(obj) is torch.Tensor
                or isinstance(obj, torch.nn.Parameter)
            ):
                nbytes = obj.numel() * obj.element_size()
                if obj.device.type == "cuda":
                    idx = obj.device.index or 0
                    cuda_need[idx] = cuda_need.get(idx, 0) + nbytes
                else:
                    cpu_bytes += nbytes
                tensor_paths.append(p)
            else:
                # ShardedTensor / quantized / arbitrary object: staging
                # must complete before the app may mutate it
                return False
        if cpu_bytes > knobs.get_shadow_cpu_max_bytes():
            return False
        if mode == "auto":
            for idx, need in cuda_need.items():
                try:
                    free, _total = torch.cuda.mem_get_info(idx)
                except Exception:
                    return False
                if free < int(need * 1.25):
                    logger.info(
                        "async shadow disabled: device %d has %.1f GB free "
                        "but the shadow needs %.1f GB",
                        idx, free / 1e9, need / 1e9,
                    )
                    return False
        # clone, preserving aliasing (tied weights -> one shared clone).
        # DTensors are rebuilt with from_local around a plain local clone:
        # DTensor.clone() dispatches through the tensor-parallel op layer
        # per call and cost ~0.6 ms x hundreds of params in the stall
        # window; from_local with explicit shape/stride is metadata-only.
        clones: Dict[Any, Any] = {}
        for p in tensor_paths:
            obj = flattened[p]
            is_dt = DTensor and isinstance(obj, DTensor)
            local = obj.to_local() if is_dt else obj
            key = (
                local.data_ptr(),
                local.dtype,
                tuple(local.shape),
                tuple(local.stride()),
                str(local.device),
            )
            if key not in clones:
                if is_dt:
                    clones[key] = DTensor.from_local(
                        local.detach().clone(),
                        obj.device_mesh,
                        obj.placements,
                        run_check=False,
                        shape=obj.shape,
                        stride=obj.stride(),
                    )
                else:
                    clones[key] = obj.detach().clone()
            flattened[p] = clones[key]
        for idx in cuda_need:
            torch.cuda.synchronize(idx)
        return True

    @classmethod
    def _partition_replicated(
        cls,
        manifest: Manifest,
        write_reqs: List[WriteReq],
        req_to_logical: Dict[str, str],
        replicated_paths: Set[str],
        pg_wrapper: PGWrapper,
    ) -> List[WriteReq]:
        rank = pg_wrapper.get_rank()
        items: List[PartitionItem] = []
        non_replicated_bytes = 0
        for req in write_reqs:
            logical = req_to_logical.get(req.path)
            # only plain replicated-marked entries participate: DTensor /
            # ShardedTensor writes are already deduplicated by their
            # preparers (replica-set round robin), and reassigning them
            # here would drop payloads whose chosen writer has the only
            # write request
            if logical in replicated_paths and getattr(
                manifest[logical], "replicated", False
            ):
                items.append(
                    PartitionItem(
                        req_path=req.path,
                        nbytes=req.stager.get_staging_cost_bytes(),
                    )
                )
            else:
                non_replicated_bytes += req.stager.get_staging_cost_bytes()

        assignment = partition_write_reqs(items, non_replicated_bytes, pg_wrapper)

        kept_reqs: List[WriteReq] = []
        dropped_logical: Dict[str, Set[str]] = {}
        for req in write_reqs:
            writer = assignment.get(req.path)
            if writer is None or writer == rank:
                kept_reqs.append(req)
            else:
                logical = req_to_logical[req.path]
                dropped_logical.setdefault(logical, set()).add(req.path)

        # a rank's manifest only claims replicated payloads it writes.
        # Tied-weight aliases SHARE the dropped entry object under other
        # logical paths — drop/replace every path holding the identical
        # object, or stale standalone locations (which the writer's batcher
        # relocates away from) would survive in this rank's manifest.
        for logical, dropped in dropped_logical.items():
            entry = manifest[logical]
            alias_paths = [p for p, e in manifest.items() if e is entry]
            if isinstance(entry, ChunkedTensorEntry):
                kept_chunks = [
                    c for c in entry.chunks if c.tensor.location not in dropped
                ]
                if kept_chunks:
                    new_entry = dataclass_replace(entry, chunks=kept_chunks)
                    for p in alias_paths:
                        manifest[p] = new_entry
                else:
                    for p in alias_paths:
                        del manifest[p]
            else:
                for p in alias_paths:
                    del manifest[p]
        return kept_reqs

    @classmethod
    def _commit(
        cls,
        storage: StoragePlugin,
        metadata: SnapshotMetadata,
        pg_wrapper: PGWrapper,
    ) -> None:
        pg_wrapper.barrier()
        if pg_wrapper.get_rank() == 0:
            storage.sync_write(
                WriteIO(
                    path=METADATA_FILENAME,
                    buf=metadata.to_json_str().encode("utf-8"),
                )
            )
        pg_wrapper.barrier()

    # -- load ---------------------------------------------------------------

    def restore(self, app_state: AppState, strict: Optional[bool] = None) -> None:
        """In-place restore. ``strict`` (when not None) is forwarded to
        load_state_dict for statefuls that accept it (nn.Module etc.), so
        partially-matching snapshots can restore with strict=False."""
        torch._C._log_api_usage_once("torchsnapshot_amd.Snapshot.restore")
        self._validate_app_state(app_state)
        pg_wrapper = PGWrapper(self.pg)
        event_meta = {
            "id": uuid.uuid4().hex,
            "rank": pg_wrapper.get_rank(),
            "api": "restore",
        }
        log_event(Event("restore_start", dict(event_meta)))
        try:
            storage = url_to_storage_plugin(self.path, self._storage_options)
            try:
                metadata = self.metadata
                rank_manifest, payload_entries = get_manifest_for_rank(
                    metadata, pg_wrapper.get_rank()
                )
                from . import integrity

                checksums = (
                    integrity.load_checksums(storage, metadata.world_size)
                    if integrity.verification_enabled()
                    else None
                )
                all_keys = self._gather_keys(app_state, pg_wrapper)
                # RNG states restore last so nothing after perturbs them
                all_keys.sort(
                    key=lambda k: isinstance(app_state.get(k), RNGState)
                )
                for key in all_keys:
                    self._load_stateful(
                        key=key,
                        stateful=app_state.get(key),
                        rank_manifest=rank_manifest,
                        payload_entries=payload_entries,
                        storage=storage,
                        pg_wrapper=pg_wrapper,
                        strict=strict,
                        checksums=checksums,
                    )
                    pg_wrapper.barrier()
            finally:
                storage.sync_close()
            log_event(Event("restore_end", {**event_meta, "success": True}))
        except Exception:
            log_event(Event("restore_end", {**event_meta, "success": False}))
            raise

    def _load_stateful(
        self,
        key: str,
        stateful: Optional[Stateful],
        rank_manifest: Manifest,
        payload_entries: Dict[str, Entry],
        storage: StoragePlugin,
        pg_wrapper: PGWrapper,
        strict: Optional[bool] = None,
        checksums: Optional[Dict[str, str]] = None,
    ) -> None:
        if stateful is None:
            return
        # in-place targets: load straight into the tensors the stateful
        # already allocated (halves peak memory)
        sd = stateful.state_dict()
        _, flattened_tgt = flatten(sd, prefix=key)

        prefix = f"{key}/"
        # deep-copied: elasticity handling may prune entries/container keys,
        # and the cached metadata must stay pristine
        import copy as _copy

        selected = {
            p: e
            for p, e in payload_entries.items()
            if p == key or p.startswith(prefix)
        }
        sub_manifest = {
            p: _copy.deepcopy(e)
            for p, e in rank_manifest.items()
            if p == key or p.startswith(prefix)
        }
        if not sub_manifest and not selected:
            raise RuntimeError(
                f"no entries found in snapshot for app-state key '{key}'"
            )
        handle_sharded_tensor_elasticity(sub_manifest, selected, flattened_tgt)

        read_reqs: List[ReadReq] = []
        futs: Dict[str, Any] = {}
        for p, entry in selected.items():
            obj_out = flattened_tgt.get(p)
            rr, fut = prepare_read(entry, obj_out)
            read_reqs.extend(rr)
            futs[p] = fut
        read_reqs = _batch_reads(read_reqs)
        budget = get_process_memory_budget_bytes(pg_wrapper)
        sync_execute_read_reqs(
            read_reqs, storage, budget, rank=pg_wrapper.get_rank(),
            checksums=checksums,
        )
        values = {p: f.obj for p, f in futs.items()}
        state_dict_to_load = inflate(sub_manifest, values, prefix=key)
        if strict is not None:
            try:
                stateful.load_state_dict(state_dict_to_load, strict=strict)
                return
            except TypeError:
                pass  # stateful doesn't accept strict
        stateful.load_state_dict(state_dict_to_load)

    def get_state_dict_for_key(self, key: str) -> Dict[str, Any]:
        """Load and return the state dict saved for ``key`` without an
        in-place target."""
        pg_wrapper = PGWrapper(self.pg)
        storage = url_to_storage_plugin(self.path, self._storage_options)
        try:
            rank_manifest, payload_entries = get_manifest_for_rank(
                self.metadata, pg_wrapper.get_rank()
            )
            prefix = f"{key}/"
            selected = {
                p: e
                for p, e in payload_entries.items()
                if p == key or p.startswith(prefix)
            }
            sub_manifest = {
                p: e
                for p, e in rank_manifest.items()
                if p == key or p.startswith(prefix)
            }
            if not sub_manifest and not selected:
                raise RuntimeError(f"key '{key}' not found in snapshot")
            read_reqs: List[ReadReq] = []
            futs: Dict[str, Any] = {}
            for p, entry in selected.items():
                rr, fut = prepare_read(entry, None)
                read_reqs.extend(rr)
                futs[p] = fut
            read_reqs = _batch_reads(read_reqs)
            budget = get_process_memory_budget_bytes(pg_wrapper)
            from . import integrity

            checksums = (
                integrity.load_checksums(storage, self.metadata.world_size)
                if integrity.verification_enabled()
                else None
            )
            sync_execute_read_reqs(
                read_reqs,
                storage,
                budget,
                rank=pg_wrapper.get_rank(),
                checksums=checksums,
            )
            values = {p: f.obj for p, f in futs.items()}
            return inflate(sub_manifest, values, prefix=key)
        finally:
            storage.sync_close()

    def read_object(
        self,
        path: str,
        obj_out: Optional[Any] = None,
        memory_budget_bytes: Optional[int] = None,
    ) -> Any:
        """Random access: load one object by its global manifest path
        ("<rank>/<logical_path>"). ``memory_budget_bytes`` bounds peak host
        memory via tiled byte-range reads."""
        torch._C._log_api_usage_once("torchsnapshot_amd.Snapshot.read_object")
        event_meta = {"id": uuid.uuid4().hex, "api": "read_object", "path": path}
        log_event(Event("read_object_start", dict(event_meta)))
        try:
            result = self._read_object_impl(path, obj_out, memory_budget_bytes)
            log_event(Event("read_object_end", {**event_meta, "success": True}))
            return result
        except Exception:
            log_event(Event("read_object_end", {**event_meta, "success": False}))
            raise

    def _read_object_impl(
        self,
        path: str,
        obj_out: Optional[Any],
        memory_budget_bytes: Optional[int],
    ) -> Any:
        rank_str, _, logical_path = path.partition("/")
        try:
            rank = int(rank_str)
        except ValueError:
            raise ValueError(
                f"read_object path must start with a rank, got {path!r}"
            ) from None
        _, payload_entries = get_manifest_for_rank(self.metadata, rank)
        if logical_path not in payload_entries:
            raise ValueError(
                f"path '{path}' does not exist in the snapshot (no entry "
                f"'{logical_path}' for rank {rank})"
            )
        entry = payload_entries[logical_path]
        if isinstance(entry, PrimitiveEntry):
            return entry.get_value()
        storage = url_to_storage_plugin(self.path, self._storage_options)
        try:
            read_reqs, fut = prepare_read(
                entry, obj_out, buffer_size_limit_bytes=memory_budget_bytes
            )
            read_reqs = _batch_reads(read_reqs)
            budget = memory_budget_bytes or (32 * 1024**3)
            from . import integrity

            checksums = (
                integrity.load_checksums(storage, self.metadata.world_size)
                if integrity.verification_enabled()
                else None
            )
            sync_execute_read_reqs(
                read_reqs, storage, budget, rank=0, checksums=checksums
            )
            return fut.obj
        finally:
            storage.sync_close()

    # -- shared helpers -----------------------------------------------------

    @staticmethod
    def _validate_app_state(app_state: AppState) -> None:
        if not isinstance(app_state, dict):
            raise TypeError(
                f"app_state must be Dict[str, Stateful], got {type(app_state)}"
            )
        for key, value in app_state.items():
            if not isinstance(key, str) or "/" in key:
                raise ValueError(
                    f"app_state key {key!r} is invalid: keys must be strings "
                    "without '/'"
                )
            if not isinstance(value, Stateful):
                raise TypeError(
                    f"app_state['{key}'] ({type(value).__name__}) does not "
                    "implement state_dict()/load_state_dict()"
                )

    @classmethod
    def _coalesce_path_and_replicated(
        cls,
        path: str,
        app_state: AppState,
        replicated: List[str],
        pg_wrapper: PGWrapper,
    ) -> Tuple[str, List[str], str]:
        # all ranks must agree on the snapshot path: rank 0 wins. A fresh
        # uid rides along so the async-commit barrier gets a unique store
        # namespace per snapshot (stale arrive/depart counters or a prior
        # failure's error_flag under the same path must not leak in).
        obj_list: List[Any] = [path, uuid.uuid4().hex]
        pg_wrapper.broadcast_object_list(obj_list, src=0)
        if obj_list[0] != path:
            logger.warning(
                "rank %d: snapshot path %r differs from rank 0's %r; using "
                "rank 0's",
                pg_wrapper.get_rank(),
                path,
                obj_list[0],
            )
        path, uid = obj_list[0], obj_list[1]

        replicated = list(replicated) + cls._infer_replicated(app_state)
        # a pattern counts only if every rank requested it
        gathered: List[Optional[List[str]]] = [None] * pg_wrapper.get_world_size()
        pg_wrapper.all_gather_object(gathered, sorted(set(replicated)))
        common = set(gathered[0] or [])
        for lst in gathered[1:]:
            common &= set(lst or [])
        dropped = set(replicated) - common
        if dropped:
            logger.warning(
                "replicated patterns %s were not requested on all ranks; "
                "ignoring them",
                sorted(dropped),
            )
        return path, sorted(common), uid

    @staticmethod
    def _infer_replicated(app_state: AppState) -> List[str]:
        """DDP modules are replicated by construction: mark their whole
        subtree (minus parameters_to_ignore)."""
        patterns: List[str] = []
        try:
            from torch.nn.parallel import DistributedDataParallel as DDP
        except ImportError:
            return patterns
        for key, stateful in app_state.items():
            if not isinstance(stateful, DDP):
                continue
            ignored = set(
                getattr(stateful, "parameters_to_ignore", None) or []
            )
            if not ignored:
                patterns.append(f"{key}/**")
            else:
                for name in stateful.state_dict().keys():
                    if name not in ignored:
                        patterns.append(f"{key}/{name}")
        return patterns

    @classmethod
    def _calculate_replicated_entries(
        cls,
        flattened: Flattened,
        replicated: List[str],
        pg_wrapper: PGWrapper,
    ) -> Set[str]:
        if not replicated:
            return set()
        candidates = sorted(
            p
            for p in flattened.keys()
            if any(glob_match(p, pat) for pat in replicated)
        )
        # a path is replicated only if it matched on EVERY rank
        gathered: List[Optional[List[str]]] = [None] * pg_wrapper.get_world_size()
        pg_wrapper.all_gather_object(gathered, candidates)
        common = set(gathered[0] or [])
        for lst in gathered[1:]:
            common &= set(lst or [])
        return common

    @staticmethod
    def _gather_keys(app_state: AppState, pg_wrapper: PGWrapper) -> List[str]:
        gathered: List[Optional[List[str]]] = [None] * pg_wrapper.get_world_size()
        pg_wrapper.all_gather_object(gathered, sorted(app_state.keys()))
        union: Set[str] = set()
        for lst in gathered:
            union |= set(lst or [])
        return sorted(union)

    @staticmethod
    def _gather_manifest(manifest: Manifest, pg_wrapper: PGWrapper) -> Manifest:
        local = {p: e.to_dict() for p, e in manifest.items()}
        gathered: List[Optional[Dict[str, Dict[str, Any]]]] = [
            None
        ] * pg_wrapper.get_world_size()
        pg_wrapper.all_gather_object(gathered, local)
        from .manifest import entry_from_dict

        global_manifest: Manifest = {}
        for rank, rank_manifest in enumerate(gathered):
            for p, d in (rank_manifest or {}).items():
                global_manifest[f"{rank}/{p}"] = entry_from_dict(d)
        return global_manifest


def _batch(write_reqs: List[WriteReq], rank: int = 0) -> List[WriteReq]:
    from .batcher import batch_write_requests

    return batch_write_requests(write_reqs, rank)


def _batch_reads(read_reqs: List[ReadReq]) -> List[ReadReq]:
    from .batcher import batch_read_requests

    return batch_read_requests(read_reqs)


# ---------------------------------------------------------------------------
# PendingSnapshot
# ---------------------------------------------------------------------------


class PendingSnapshot:
    """Returned by async_take: storage I/O drains on a background thread;
    the commit is coordinated with a store barrier (no collectives off the
    main thread)."""

    def __init__(
        self,
        path: str,
        pending_io_work: PendingIOWork,
        pg_wrapper: PGWrapper,
        metadata: SnapshotMetadata,
        storage: StoragePlugin,
        barrier: LinearBarrier,
        storage_options: Optional[Dict[str, Any]],
        event_meta: Optional[Dict[str, Any]] = None,
    ) -> None:
        self.path = path
        self._pending_io_work = pending_io_work
        # True when the snapshot's sources were shadow-cloned: the app
        # was free to mutate its state the moment async_take returned
        self.sources_immutable = pending_io_work.sources_immutable
        self._pg_wrapper = pg_wrapper
        self._metadata = metadata
        self._storage = storage
        self._storage_options = storage_options
        self._event_meta = event_meta or {}
        self._barrier = barrier
        self._exc: Optional[BaseException] = None
        self._done_event = threading.Event()
        self._thread = threading.Thread(
            target=self._complete, name="tsamd-async-commit", daemon=True
        )
        self._thread.start()

    def _complete(self) -> None:
        try:
            self._pending_io_work.complete()
            from .integrity import write_checksum_file

            write_checksum_file(
                self._storage,
                self._pg_wrapper.get_rank(),
                self._pending_io_work.checksums,
            )
            self._barrier.arrive()
            if self._pg_wrapper.get_rank() == 0:
                self._storage.sync_write(
                    WriteIO(
                        path=METADATA_FILENAME,
                        buf=self._metadata.to_json_str().encode("utf-8"),
                    )
                )
            self._barrier.depart()
            log_event(
                Event("async_take_end", {**self._event_meta, "success": True})
            )
        except BaseException as e:  # noqa: B036
            self._exc = e
            try:
                self._barrier.report_error(e)
            except Exception:
                logger.exception("failed to report async-commit error")
            log_event(
                Event("async_take_end", {**self._event_meta, "success": False})
            )
        finally:
            try:
                self._storage.sync_close()
            except Exception:
                logger.exception("failed to close storage after async take")
            self._done_event.set()

    def wait(self) -> Snapshot:
        self._thread.join()
        if self._exc is not None:
            raise RuntimeError(
                "async snapshot failed; no metadata was committed"
            ) from self._exc
        snapshot = Snapshot(
            self.path, self._pg_wrapper.pg, self._storage_options
        )
        snapshot._metadata = self._metadata
        return snapshot

    def done(self) -> bool:
        return self._done_event.is_set()
