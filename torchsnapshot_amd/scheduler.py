"""Pipelined execution of write/read requests under a host-memory budget.

Design (differs from the reference's nested-event-loop scheduler,
torchsnapshot/scheduler.py:222-446): every pipeline runs on a dedicated
background thread with its own asyncio event loop. One coroutine per
request drives stage -> write (or read -> consume); a budget condition
variable bounds the total bytes of in-flight buffers, and a semaphore
bounds storage concurrency. Because the caller thread never runs the loop,
the same machinery serves sync take (wait for everything), async_take
(wait for staging only — the thread keeps draining storage I/O), and
nested/Jupyter callers without re-entrant-loop tricks.

Collectives never run on the pipeline thread (RCCL/process-group calls stay
on the caller thread), matching the constraint the reference documents at
snapshot.py:1010.
"""

from __future__ import annotations

import asyncio
import logging
import socket
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from dataclasses import dataclass, field
from typing import Awaitable, Callable, List, Optional, TypeVar

import psutil

from . import knobs
from .io_types import ReadIO, ReadReq, StageContext, StoragePlugin, WriteIO, WriteReq
from .pg_wrapper import PGWrapper
from .roctx import roctx_range
from . import integrity

logger = logging.getLogger(__name__)

T = TypeVar("T")

_MAX_PER_RANK_MEMORY_BUDGET_BYTES = 32 * 1024 * 1024 * 1024
_AVAILABLE_MEMORY_FRACTION = 0.6


def run_coro_sync(coro: Awaitable[T]) -> T:
    """Run a coroutine to completion from sync code, safely even when the
    caller is already inside a running event loop (e.g. Jupyter)."""
    try:
        asyncio.get_running_loop()
    except RuntimeError:
        return asyncio.run(coro)  # type: ignore[arg-type]
    with ThreadPoolExecutor(max_workers=1) as ex:
        return ex.submit(asyncio.run, coro).result()  # type: ignore[arg-type]


def get_local_world_size(pg: PGWrapper) -> int:
    """Number of ranks on this host (divides the host-memory budget)."""
    hostnames: List[Optional[str]] = [None] * pg.get_world_size()
    pg.all_gather_object(hostnames, socket.gethostname())
    return hostnames.count(socket.gethostname())


def get_process_memory_budget_bytes(pg: PGWrapper) -> int:
    override = knobs.get_memory_budget_override_bytes()
    if override is not None:
        logger.info("Manual memory budget: %d bytes", override)
        return override
    available = psutil.virtual_memory().available
    budget = int(
        available * _AVAILABLE_MEMORY_FRACTION / max(get_local_world_size(pg), 1)
    )
    return min(budget, _MAX_PER_RANK_MEMORY_BUDGET_BYTES)


@dataclass
class ExecutionStats:
    total_reqs: int = 0
    staged_reqs: int = 0
    done_reqs: int = 0
    staged_bytes: int = 0
    io_bytes: int = 0
    # cumulative per-phase busy time across requests (can exceed wall
    # time with concurrency); enable logging with TSAMD_TIMING=1
    stage_s: float = 0.0
    io_s: float = 0.0
    consume_s: float = 0.0
    begin_ts: float = field(default_factory=time.monotonic)
    staged_ts: Optional[float] = None
    end_ts: Optional[float] = None

    def throughput_bytes_per_sec(self) -> float:
        end = self.end_ts or time.monotonic()
        dur = max(end - self.begin_ts, 1e-9)
        return self.io_bytes / dur


class _Budget:
    """Async counter of in-flight buffer bytes with an escape hatch: a
    request costing more than the whole budget may run when the pipeline is
    otherwise empty (so oversized items make progress instead of
    deadlocking)."""

    def __init__(self, limit: int) -> None:
        self.limit = limit
        self.in_use = 0
        self.cond = asyncio.Condition()

    async def acquire(self, cost: int) -> None:
        async with self.cond:
            while not (
                self.in_use + cost <= self.limit
                or (self.in_use == 0 and cost > self.limit)
            ):
                await self.cond.wait()
            self.in_use += cost

    async def release(self, cost: int) -> None:
        async with self.cond:
            self.in_use -= cost
            self.cond.notify_all()


class PendingIOWork:
    """Handle to an in-flight pipeline running on its own thread."""

    def __init__(
        self,
        thread: threading.Thread,
        staged_event: threading.Event,
        done_event: threading.Event,
        stats: ExecutionStats,
        exc_holder: List[BaseException],
        checksums: Optional[dict] = None,
    ) -> None:
        self._thread = thread
        self._staged_event = staged_event
        self._done_event = done_event
        self.stats = stats
        self._exc_holder = exc_holder
        # {payload_path: xxh3 hex}, filled when TSAMD_CHECKSUM=1
        self.checksums = checksums if checksums is not None else {}
        # True when every source was shadow-cloned before the pipeline
        # started (async saves): the app may mutate its state without
        # waiting for staging
        self.sources_immutable = False

    def _maybe_raise(self) -> None:
        if self._exc_holder:
            raise self._exc_holder[0]

    def wait_staged(self) -> None:
        """Block until all buffers are staged in host memory (storage I/O may
        still be in flight). After this, the source tensors are safe to
        mutate."""
        self._staged_event.wait()
        if not self._done_event.is_set():
            # staging done; errors so far would have set done as well
            if self._exc_holder:
                self._thread.join()
                self._maybe_raise()
        else:
            self._maybe_raise()

    def complete(self) -> None:
        """Block until all storage I/O finished; re-raise pipeline errors."""
        self._thread.join()
        self._maybe_raise()
        import os

        if os.environ.get("TSAMD_TIMING") and self.stats.staged_bytes:
            s = self.stats
            wall = (s.end_ts or 0) - s.begin_ts
            print(
                f"[tsamd timing] pipeline: wall={wall:.2f}s reqs={s.total_reqs} "
                f"staged={s.staged_bytes/1e9:.2f}GB stage_busy={s.stage_s:.2f}s "
                f"io_busy={s.io_s:.2f}s consume_busy={s.consume_s:.2f}s"
            )

    def done(self) -> bool:
        return self._done_event.is_set()


def _spawn_pipeline(
    main: Callable[[], Awaitable[None]],
    stats: ExecutionStats,
    staged_event: threading.Event,
    done_event: threading.Event,
    checksums: Optional[dict] = None,
) -> PendingIOWork:
    exc_holder: List[BaseException] = []

    def runner() -> None:
        try:
            asyncio.run(main())
        except BaseException as e:  # noqa: B036
            exc_holder.append(e)
        finally:
            stats.end_ts = time.monotonic()
            staged_event.set()
            done_event.set()

    thread = threading.Thread(
        target=runner, name="tsamd-io-pipeline", daemon=True
    )
    thread.start()
    return PendingIOWork(
        thread, staged_event, done_event, stats, exc_holder, checksums
    )


# ---------------------------------------------------------------------------
# write pipeline
# ---------------------------------------------------------------------------


def execute_write_reqs(
    write_reqs: List[WriteReq],
    storage: StoragePlugin,
    memory_budget_bytes: int,
    rank: int,
    is_async: bool = False,
) -> PendingIOWork:
    stats = ExecutionStats(total_reqs=len(write_reqs))
    staged_event = threading.Event()
    done_event = threading.Event()
    checksums: dict = {}
    do_checksum = integrity.checksumming_enabled()

    # Largest first: big buffers claim budget early, small ones fill gaps.
    ordered = sorted(
        write_reqs, key=lambda r: r.stager.get_staging_cost_bytes(), reverse=True
    )

    async def main() -> None:
        budget = _Budget(memory_budget_bytes)
        io_sem = asyncio.Semaphore(knobs.get_max_io_concurrency())
        staging_sem = asyncio.Semaphore(knobs.get_num_staging_threads() * 2)
        executor = ThreadPoolExecutor(
            max_workers=knobs.get_num_staging_threads(),
            thread_name_prefix="tsamd-stage",
        )
        ctx = StageContext(executor=executor, is_async=is_async)
        staged_remaining = len(ordered)
        all_staged = asyncio.Event()
        if staged_remaining == 0:
            all_staged.set()

        async def handle(req: WriteReq) -> None:
            nonlocal staged_remaining
            cost = req.stager.get_staging_cost_bytes()
            await budget.acquire(cost)
            try:
                async with staging_sem:
                    t0 = time.monotonic()
                    with roctx_range(f"tsamd:stage:{req.path}"):
                        buf = await req.stager.stage_buffer(ctx)
                    stats.stage_s += time.monotonic() - t0
                nbytes = memoryview(buf).nbytes
                if do_checksum:
                    pre = getattr(req.stager, "precomputed_checksum", None)
                    if pre is not None:
                        checksums[req.path] = pre
                    else:
                        # psum64 for host-staged payloads too (native
                        # single-pass hash, ~14 GB/s): every file shares
                        # one algorithm, so byte-range/tiled reads of ANY
                        # payload are verifiable on restore
                        checksums[req.path] = await asyncio.get_running_loop(
                        ).run_in_executor(
                            executor, integrity.psum64_hexdigest, buf
                        )
                    # per-member values so byte-range/merged-span reads of
                    # batched slabs are verifiable on restore
                    for s, e, v in (
                        getattr(req.stager, "member_checksums", None) or []
                    ):
                        checksums[integrity.member_key(req.path, s, e)] = v
                    # file length, so TILED byte-range reads of this file
                    # can be verified once their union covers the file
                    checksums[integrity.len_key(req.path)] = str(nbytes)
                stats.staged_reqs += 1
                stats.staged_bytes += nbytes
                staged_remaining -= 1
                if staged_remaining == 0:
                    all_staged.set()
                async with io_sem:
                    t0 = time.monotonic()
                    with roctx_range(f"tsamd:write:{req.path}"):
                        await storage.write(WriteIO(path=req.path, buf=buf))
                    stats.io_s += time.monotonic() - t0
                stats.io_bytes += nbytes
                stats.done_reqs += 1
            finally:
                # release pooled staging buffers on success AND failure so
                # a failed snapshot doesn't starve the pinned pool
                try:
                    req.stager.release_buffer()
                except Exception:
                    logger.exception("release_buffer failed for %s", req.path)
                await budget.release(cost)

        tasks = [asyncio.create_task(handle(r)) for r in ordered]
        reporter = asyncio.create_task(_report_progress(stats, rank, "write"))

        async def signal_staged() -> None:
            await all_staged.wait()
            stats.staged_ts = time.monotonic()
            staged_event.set()

        signaler = asyncio.create_task(signal_staged())
        try:
            results = await asyncio.gather(*tasks, return_exceptions=True)
            errors = [r for r in results if isinstance(r, BaseException)]
            if errors:
                raise errors[0]
        finally:
            signaler.cancel()
            reporter.cancel()
            executor.shutdown(wait=False)
            try:
                # release per-event-loop resources (network backends keep
                # one session per loop; this loop dies with the pipeline)
                await storage.close_for_loop()
            except Exception:
                logger.exception("storage close at pipeline end failed")

    return _spawn_pipeline(main, stats, staged_event, done_event, checksums)


def sync_execute_write_reqs(
    write_reqs: List[WriteReq],
    storage: StoragePlugin,
    memory_budget_bytes: int,
    rank: int,
) -> ExecutionStats:
    pending = execute_write_reqs(
        write_reqs, storage, memory_budget_bytes, rank=rank, is_async=False
    )
    pending.complete()
    import os

    if os.environ.get("TSAMD_TIMING"):
        s = pending.stats
        wall = (s.end_ts or 0) - s.begin_ts
        print(
            f"[tsamd timing] write: wall={wall:.2f}s reqs={s.total_reqs} "
            f"bytes={s.io_bytes/1e9:.2f}GB stage_busy={s.stage_s:.2f}s "
            f"io_busy={s.io_s:.2f}s"
        )
    if rank == 0 and pending.stats.io_bytes:
        logger.info(
            "Wrote %.1f MB in %.2fs (%.2f GB/s)",
            pending.stats.io_bytes / 1e6,
            (pending.stats.end_ts or 0) - pending.stats.begin_ts,
            pending.stats.throughput_bytes_per_sec() / 1e9,
        )
    return pending.stats


# ---------------------------------------------------------------------------
# read pipeline
# ---------------------------------------------------------------------------


def execute_read_reqs(
    read_reqs: List[ReadReq],
    storage: StoragePlugin,
    memory_budget_bytes: int,
    rank: int,
    checksums: Optional[dict] = None,
) -> PendingIOWork:
    stats = ExecutionStats(total_reqs=len(read_reqs))
    staged_event = threading.Event()
    done_event = threading.Event()

    ordered = sorted(
        read_reqs,
        key=lambda r: r.consumer.get_consuming_cost_bytes(),
        reverse=True,
    )

    async def main() -> None:
        budget = _Budget(memory_budget_bytes)
        io_sem = asyncio.Semaphore(knobs.get_max_io_concurrency())
        executor = ThreadPoolExecutor(
            max_workers=knobs.get_num_staging_threads(),
            thread_name_prefix="tsamd-consume",
        )
        ctx = StageContext(executor=executor)
        # path -> [accumulated psum64, [(start, end), ...]] for tiled reads
        partial_sums: Dict[str, list] = {}

        async def handle(req: ReadReq) -> None:
            cost = req.consumer.get_consuming_cost_bytes()
            await budget.acquire(cost)
            try:
                read_io = ReadIO(
                    path=req.path,
                    byte_range=req.byte_range,
                    buf_alloc=req.buf_alloc,
                )
                async with io_sem:
                    t0 = time.monotonic()
                    await storage.read(read_io)
                    stats.io_s += time.monotonic() - t0
                buf = read_io.buf
                stats.io_bytes += memoryview(buf).nbytes
                if checksums:
                    loop = asyncio.get_running_loop()
                    # device-bound consumers verify the bytes ON the GPU
                    # right after their H2D copy (HBM-speed hash; a CPU
                    # hash here would bottleneck a warm 50 GB/s restore)
                    dev_exp = None
                    try:
                        if req.consumer.will_verify_on_device():
                            if req.byte_range is None:
                                want = checksums.get(req.path, "")
                                if want.startswith("psum64:"):
                                    dev_exp = (int(want[7:], 16), 0)
                            else:
                                s0, e0 = req.byte_range
                                total = integrity.expected_span_psum(
                                    req.path, req.byte_range, checksums
                                )
                                if total is not None and s0 % 8 == 0:
                                    dev_exp = (total, s0 // 8)
                    except AttributeError:
                        pass
                    if dev_exp is not None:
                        req.consumer.expected_psum = dev_exp
                    elif req.byte_range is None:
                        await loop.run_in_executor(
                            executor, integrity.verify_buffer, req.path, buf,
                            checksums,
                        )
                    else:
                        verified = await loop.run_in_executor(
                            executor, integrity.verify_ranged_buffer,
                            req.path, buf, req.byte_range, checksums,
                        )
                        s, e = req.byte_range
                        whole = checksums.get(req.path, "")
                        if (
                            not verified
                            and whole.startswith("psum64:")
                            and s % 8 == 0
                        ):
                            # tiled read of an unbatched file: accumulate
                            # the partial sum; once the tiles cover the
                            # whole file, the total is checked below
                            val = await loop.run_in_executor(
                                executor, integrity.psum64_value, buf, s // 8
                            )
                            rec = partial_sums.setdefault(
                                req.path, [0, []]
                            )
                            rec[0] = (rec[0] + val) % (1 << 64)
                            rec[1].append((s, e))
                t0 = time.monotonic()
                await req.consumer.consume_buffer(ctx, buf)
                stats.consume_s += time.monotonic() - t0
                stats.done_reqs += 1
            finally:
                req.consumer.close()
                await budget.release(cost)

        tasks = [asyncio.create_task(handle(r)) for r in ordered]
        reporter = asyncio.create_task(_report_progress(stats, rank, "read"))
        try:
            results = await asyncio.gather(*tasks, return_exceptions=True)
            errors = [r for r in results if isinstance(r, BaseException)]
            if errors:
                raise errors[0]
            # tiled-read verification: files whose byte-range tiles ended
            # up covering the whole file get their accumulated psum64
            # checked against the recorded whole-file value
            for path, (total, ranges) in partial_sums.items():
                want = checksums.get(path) if checksums else None
                want_len = checksums.get(integrity.len_key(path)) if checksums else None
                if not want or not want_len:
                    continue
                ranges.sort()
                covered = 0
                for s, e in ranges:
                    if s != covered:
                        covered = -1
                        break
                    covered = e
                if covered != int(want_len):
                    continue  # partial coverage: cannot verify
                got = "psum64:" + format(total, "016x")
                if got != want:
                    raise RuntimeError(
                        f"checksum mismatch for tiled read of '{path}': "
                        f"snapshot recorded {want}, read back {got} — the "
                        "file is corrupted or was modified after the "
                        "snapshot was committed"
                    )
        finally:
            reporter.cancel()
            executor.shutdown(wait=False)
            try:
                await storage.close_for_loop()
            except Exception:
                logger.exception("storage close at pipeline end failed")

    return _spawn_pipeline(main, stats, staged_event, done_event)


def sync_execute_read_reqs(
    read_reqs: List[ReadReq],
    storage: StoragePlugin,
    memory_budget_bytes: int,
    rank: int,
    checksums: Optional[dict] = None,
) -> ExecutionStats:
    pending = execute_read_reqs(
        read_reqs, storage, memory_budget_bytes, rank, checksums
    )
    pending.complete()
    import os

    if os.environ.get("TSAMD_TIMING"):
        s = pending.stats
        wall = (s.end_ts or 0) - s.begin_ts
        print(
            f"[tsamd timing] read: wall={wall:.2f}s reqs={s.total_reqs} "
            f"bytes={s.io_bytes/1e9:.2f}GB io_busy={s.io_s:.2f}s "
            f"consume_busy={s.consume_s:.2f}s"
        )
    if rank == 0 and pending.stats.io_bytes:
        logger.info(
            "Read %.1f MB in %.2fs (%.2f GB/s)",
            pending.stats.io_bytes / 1e6,
            (pending.stats.end_ts or 0) - pending.stats.begin_ts,
            pending.stats.throughput_bytes_per_sec() / 1e9,
        )
    return pending.stats


async def _report_progress(stats: ExecutionStats, rank: int, verb: str) -> None:
    if rank != 0:
        return
    interval = 10.0
    while True:
        await asyncio.sleep(interval)
        logger.info(
            "[rank %d] %s progress: %d/%d reqs, %.1f MB staged, %.1f MB io",
            rank,
            verb,
            stats.done_reqs,
            stats.total_reqs,
            stats.staged_bytes / 1e6,
            stats.io_bytes / 1e6,
        )
