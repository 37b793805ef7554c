"""Tunable knobs, overridable via environment variables.

Parity with reference torchsnapshot/knobs.py:23-132 (env-var constants +
context-manager overrides for tests), re-tuned for MI355X nodes:

- The per-GPU HBM3E pool is 288 GB and host DDR on an MI355X node is large,
  so default chunk/shard/slab sizes are bigger than the reference's
  (512 MB/512 MB/128 MB): larger slabs mean fewer, larger NVMe writes and
  fewer D2H launches over the PCIe Gen5 link (~55-60 GB/s effective).
"""

from __future__ import annotations

import contextlib
import os
from typing import Iterator

_MB = 1024 * 1024


def _env_bytes(name: str, default: int) -> int:
    val = os.environ.get(name)
    if val is None:
        return default
    return int(float(val))


def _env_int(name: str, default: int) -> int:
    val = os.environ.get(name)
    if val is None:
        return default
    return int(val)


def _env_flag(name: str) -> bool:
    return os.environ.get(name, "0") not in ("0", "", "false", "False")


# -- write-path sizing -------------------------------------------------------

def get_max_chunk_size_bytes() -> int:
    """Tensors larger than this are split into chunks for pipelined I/O."""
    return _env_bytes("TSAMD_MAX_CHUNK_SIZE_BYTES", 512 * _MB)


def get_max_shard_size_bytes() -> int:
    """Local shards of sharded tensors are subdivided to at most this size."""
    return _env_bytes("TSAMD_MAX_SHARD_SIZE_BYTES", 512 * _MB)


def get_slab_size_threshold_bytes() -> int:
    """Write requests smaller than this are packed into batched slabs."""
    return _env_bytes("TSAMD_SLAB_SIZE_THRESHOLD_BYTES", 512 * _MB)


def is_batching_disabled() -> bool:
    return _env_flag("TSAMD_DISABLE_BATCHING")


def is_partitioner_disabled() -> bool:
    return _env_flag("TSAMD_DISABLE_PARTITIONER")


# -- execution ---------------------------------------------------------------

def get_max_io_concurrency() -> int:
    """Maximum concurrent storage I/O operations per rank (32 measured
    best for both the steady-state save and the cold parallel-segment
    restore on MI355X NVMe, profiles/r02_measurements.md)."""
    return _env_int("TSAMD_MAX_PER_RANK_IO_CONCURRENCY", 32)


def get_num_staging_threads() -> int:
    """Executor threads used for CPU-side staging (GIL-releasing copies)."""
    return _env_int("TSAMD_NUM_STAGING_THREADS", 8)


def get_memory_budget_override_bytes() -> int | None:
    val = os.environ.get("TSAMD_PER_RANK_MEMORY_BUDGET_BYTES")
    return None if val is None else int(float(val))


def get_pinned_block_size_bytes() -> int:
    """Size of one pinned host staging block in the D2H ring. Must be at
    least the max chunk size so chunk stagers hit the pool, not one-off
    pinned allocations."""
    return _env_bytes(
        "TSAMD_PINNED_BLOCK_SIZE_BYTES", max(get_max_chunk_size_bytes(), 512 * _MB)
    )


def get_pinned_pool_bytes() -> int:
    """Total pinned host memory for the D2H staging ring, per rank. Sized
    so an async snapshot of a large sharded model stages without stalling:
    a quarter of host RAM divided among the local ranks, capped at 24 GB."""
    override = os.environ.get("TSAMD_PINNED_POOL_BYTES")
    if override is not None:
        return int(float(override))
    import psutil

    local_ws = int(os.environ.get("LOCAL_WORLD_SIZE", "1") or "1")
    total = psutil.virtual_memory().total
    return min(24 * 1024 * _MB, int(total * 0.35)) // max(local_ws, 1)


def get_pinned_block_count() -> int:
    """Number of pinned host staging blocks in the D2H ring."""
    override = os.environ.get("TSAMD_PINNED_BLOCK_COUNT")
    if override is not None:
        return int(override)
    return max(get_pinned_pool_bytes() // get_pinned_block_size_bytes(), 2)


def is_hip_staging_disabled() -> bool:
    """Force the torch fallback for device staging (debug only)."""
    return _env_flag("TSAMD_DISABLE_HIP_STAGING")


def get_async_shadow_mode() -> str:
    """Shadow-clone async snapshots: "auto" (default) clones device
    tensors at HBM rate before async_take returns, when free HBM allows —
    training resumes after milliseconds instead of waiting for the D2H
    staging of the whole model ("1" forces it, "0" disables)."""
    val = os.environ.get("TSAMD_ASYNC_SHADOW", "auto").lower()
    return val if val in ("auto", "1", "0") else "auto"


def get_shadow_cpu_max_bytes() -> int:
    """Cap on CPU-tensor bytes eagerly cloned in shadow mode: large host
    states clone faster in the parallel staging pipeline than on the
    caller thread, so beyond this the classic wait-for-staging path wins."""
    return _env_bytes("TSAMD_SHADOW_CPU_MAX_BYTES", 1024 * _MB)


def get_storage_write_chunk_bytes() -> int:
    """Chunk size for filesystem pwrite calls (large sequential writes)."""
    return _env_bytes("TSAMD_FS_WRITE_CHUNK_BYTES", 256 * _MB)


def get_fs_parallel_io_min_bytes() -> int:
    """Files at/above this size are READ as concurrent segments (multiple
    NVMe queues per file; closes the cold-read gap vs raw device bandwidth
    that a single-stream pread leaves). Writes stay single-stream per file
    — buffered writes serialize on the inode lock anyway."""
    return _env_bytes("TSAMD_FS_PARALLEL_IO_MIN_BYTES", 128 * _MB)


def get_fs_io_segment_bytes() -> int:
    """Segment size for parallel per-file reads."""
    return _env_bytes("TSAMD_FS_IO_SEGMENT_BYTES", 64 * _MB)


# -- elasticity --------------------------------------------------------------

def is_sharded_elasticity_root_only() -> bool:
    """If set, sharded-tensor elasticity handling only applies at state-dict
    root level (mirrors the reference's narrow-scope knob)."""
    return _env_flag("TSAMD_ENABLE_SHARDED_TENSOR_ELASTICITY_ROOT_ONLY")


# -- test overrides ----------------------------------------------------------

@contextlib.contextmanager
def override_env(name: str, value: str) -> Iterator[None]:
    prev = os.environ.get(name)
    os.environ[name] = value
    try:
        yield
    finally:
        if prev is None:
            os.environ.pop(name, None)
        else:
            os.environ[name] = prev


def override_max_chunk_size_bytes(nbytes: int):
    return override_env("TSAMD_MAX_CHUNK_SIZE_BYTES", str(nbytes))


def override_max_shard_size_bytes(nbytes: int):
    return override_env("TSAMD_MAX_SHARD_SIZE_BYTES", str(nbytes))


def override_slab_size_threshold_bytes(nbytes: int):
    return override_env("TSAMD_SLAB_SIZE_THRESHOLD_BYTES", str(nbytes))


def override_batching_disabled(disabled: bool):
    return override_env("TSAMD_DISABLE_BATCHING", "1" if disabled else "0")


def override_max_io_concurrency(n: int):
    return override_env("TSAMD_MAX_PER_RANK_IO_CONCURRENCY", str(n))
