"""Local/remote-mounted filesystem storage backend.

Design (MI355X node, local NVMe target): blocking pwrite/pread on raw fds in
a worker-thread pool — the syscalls release the GIL, so N threads drive N
NVMe queues concurrently. Writes are large, sequential, ONE stream per file
(buffered writes to one inode serialize on i_rwsem, so intra-file write
parallelism only costs cross-file concurrency — measured, see
profiles/r02_measurements.md); the batcher packs small tensors into
multi-hundred-MB slabs upstream, which is the layout NVMe likes. Reads of
files above ``TSAMD_FS_PARALLEL_IO_MIN_BYTES`` fan out into concurrent
disjoint segments with SEQUENTIAL+WILLNEED fadvise readahead — a
single-stream pread is readahead-window-bound and leaves ~20% of raw NVMe
bandwidth on a cold read. No asyncio file libraries are used (parity of
behavior with the reference's aiofiles plugin,
torchsnapshot/storage_plugins/fs.py:28-51, via a different mechanism).
"""

from __future__ import annotations

import asyncio
import os
import shutil
from concurrent.futures import ThreadPoolExecutor
from typing import List, Optional, Set

from .. import knobs
from ..io_types import ReadIO, StoragePlugin, WriteIO


def _fsync_enabled() -> bool:
    """TSAMD_FSYNC=1: fsync every payload (and its directory) so a
    committed snapshot survives power loss, not just process crashes.
    Costs raw-disk write throughput; page-cache-speed saves are the
    default (the reference's aiofiles plugin never syncs either)."""
    return os.environ.get("TSAMD_FSYNC", "0") not in ("0", "", "false")


def _as_u8_mv(buf) -> memoryview:
    mv = memoryview(buf)
    if mv.format != "B":
        mv = mv.cast("B")
    return mv


class FSStoragePlugin(StoragePlugin):
    def __init__(self, root: str, storage_options: Optional[dict] = None) -> None:
        self.root = root
        opts = storage_options or {}
        nthreads = int(opts.get("io_threads", knobs.get_max_io_concurrency()))
        self._executor = ThreadPoolExecutor(
            max_workers=nthreads, thread_name_prefix="tsamd-fs"
        )
        self._created_dirs: Set[str] = set()

    # -- helpers ------------------------------------------------------------

    def _abspath(self, path: str) -> str:
        return os.path.join(self.root, path)

    def _ensure_dir(self, dirname: str) -> None:
        if dirname in self._created_dirs:
            return
        os.makedirs(dirname, exist_ok=True)
        self._created_dirs.add(dirname)

    def _fsync_file_and_dir(self, fd: int, full: str) -> None:
        os.fsync(fd)
        # make the directory entry durable too (crash consistency: the
        # metadata commit must never be durable before its payloads)
        dfd = os.open(os.path.dirname(full), os.O_RDONLY)
        try:
            os.fsync(dfd)
        finally:
            os.close(dfd)

    def _write_sync(self, path: str, buf) -> None:
        full = self._abspath(path)
        self._ensure_dir(os.path.dirname(full))
        mv = _as_u8_mv(buf)
        fd = os.open(full, os.O_WRONLY | os.O_CREAT | os.O_TRUNC, 0o644)
        try:
            chunk = knobs.get_storage_write_chunk_bytes()
            off = 0
            total = len(mv)
            while off < total:
                off += os.pwrite(fd, mv[off : off + chunk], off)
            if _fsync_enabled():
                self._fsync_file_and_dir(fd, full)
            # note: fadvise(DONTNEED) after payload writes was measured
            # and rejected — it did not lift the sustained save rate
            # (9.50 vs 9.40 GB/s, writeback equilibrium dominates either
            # way) and it cost warm restores 3x (50 -> 16 GB/s).
            # profiles/r02_measurements.md.
        finally:
            os.close(fd)

    def _pread_segment(
        self, full: str, mv: memoryview, file_off: int
    ) -> None:
        """Read len(mv) bytes at file_off into mv with sequential-readahead
        hints; its own fd so per-fd readahead state is not shared."""
        fd = os.open(full, os.O_RDONLY)
        try:
            try:
                os.posix_fadvise(
                    fd, file_off, len(mv), os.POSIX_FADV_SEQUENTIAL
                )
                if os.environ.get("TSAMD_FS_WILLNEED", "1") not in ("0", ""):
                    os.posix_fadvise(
                        fd, file_off, len(mv), os.POSIX_FADV_WILLNEED
                    )
            except (AttributeError, OSError):
                pass
            nbytes = len(mv)
            off = 0
            while off < nbytes:
                n = os.preadv(fd, [mv[off:]], file_off + off)
                if n == 0:
                    raise EOFError(
                        f"unexpected EOF reading {full} at offset "
                        f"{file_off + off} (wanted {nbytes} from {file_off})"
                    )
                off += n
        finally:
            os.close(fd)

    def _read_sync(self, read_io: ReadIO) -> None:
        full = self._abspath(read_io.path)
        if read_io.byte_range is None:
            start, end = 0, os.path.getsize(full)
        else:
            start, end = read_io.byte_range
        nbytes = end - start
        if read_io.buf_alloc is not None:
            buf = read_io.buf_alloc(nbytes)
        else:
            buf = bytearray(nbytes)
        self._pread_segment(full, _as_u8_mv(buf), start)
        read_io.buf = buf

    @staticmethod
    def _segments(nbytes: int) -> List[tuple]:
        seg = knobs.get_fs_io_segment_bytes()
        return [(o, min(o + seg, nbytes)) for o in range(0, nbytes, seg)]

    # -- StoragePlugin ------------------------------------------------------

    async def write(self, write_io: WriteIO) -> None:
        # one sequential stream per file: ext4/xfs serialize buffered
        # writes to one inode on i_rwsem, so intra-file write parallelism
        # only steals executor threads from OTHER files (measured: 10-step
        # sustained save fell 9.9 -> 5.8 GB/s with segmented writes, and
        # durable saves didn't improve — profiles/r02_measurements.md).
        # Cross-file concurrency comes from the scheduler's many in-flight
        # write requests.
        loop = asyncio.get_running_loop()
        await loop.run_in_executor(
            self._executor, self._write_sync, write_io.path, write_io.buf
        )

    async def read(self, read_io: ReadIO) -> None:
        loop = asyncio.get_running_loop()
        full = self._abspath(read_io.path)
        if read_io.byte_range is None:
            start = 0
            end = await loop.run_in_executor(None, os.path.getsize, full)
        else:
            start, end = read_io.byte_range
        nbytes = end - start
        if nbytes < knobs.get_fs_parallel_io_min_bytes():
            await loop.run_in_executor(self._executor, self._read_sync, read_io)
            return
        # large file: concurrent segment preads (multiple NVMe queues)
        if read_io.buf_alloc is not None:
            buf = read_io.buf_alloc(nbytes)
        else:
            buf = bytearray(nbytes)
        mv = _as_u8_mv(buf)
        await asyncio.gather(
            *(
                loop.run_in_executor(
                    self._executor,
                    self._pread_segment,
                    full,
                    mv[s:e],
                    start + s,
                )
                for s, e in self._segments(nbytes)
            )
        )
        read_io.buf = buf

    async def delete(self, path: str) -> None:
        loop = asyncio.get_running_loop()
        await loop.run_in_executor(self._executor, os.remove, self._abspath(path))

    async def delete_dir(self, path: str) -> None:
        loop = asyncio.get_running_loop()
        await loop.run_in_executor(
            self._executor, shutil.rmtree, self._abspath(path)
        )

    async def close(self) -> None:
        self._executor.shutdown(wait=False)
