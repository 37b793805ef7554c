"""Core I/O abstractions shared by preparers, scheduler and storage.

Parity with reference torchsnapshot/io_types.py:24-120, re-designed around
an always-off-main-thread asyncio pipeline (see scheduler.py):

- ``BufferStager`` produces the bytes for one write request (for device
  tensors this drives the HIP staging engine: gather-pack kernel + D2H into
  pinned memory).
- ``BufferConsumer`` applies the bytes of one read request to its
  destination (H2D + strided scatter for device tensors).
- ``StoragePlugin`` moves bytes to/from a storage backend, async.
"""

from __future__ import annotations

import abc
from concurrent.futures import ThreadPoolExecutor
from dataclasses import dataclass, field
from typing import Any, Optional, Tuple

BufferType = Any  # bytes | bytearray | memoryview


@dataclass
class StageContext:
    """Resources available to stagers/consumers during pipeline execution."""

    executor: ThreadPoolExecutor
    # Per-call flag: async snapshots must defensively copy CPU tensors that
    # training might mutate after async_take returns.
    is_async: bool = False


class BufferStager(abc.ABC):
    @abc.abstractmethod
    async def stage_buffer(self, ctx: StageContext) -> BufferType:
        """Produce the serialized bytes for this write request."""

    @abc.abstractmethod
    def get_staging_cost_bytes(self) -> int:
        """Peak host-memory cost of staging this buffer (for budgeting)."""

    def release_buffer(self) -> None:
        """Called once the storage write completed; return pooled buffers."""


class BufferConsumer(abc.ABC):
    @abc.abstractmethod
    async def consume_buffer(self, ctx: StageContext, buf: BufferType) -> None:
        """Apply the read bytes to the destination object."""

    @abc.abstractmethod
    def get_consuming_cost_bytes(self) -> int:
        """Peak host-memory cost of holding + consuming this buffer."""

    def close(self) -> None:
        """Called when the pipeline is done with this request (success or
        failure); return pooled buffers here."""


@dataclass
class WriteReq:
    path: str
    stager: BufferStager
    # The TensorEntry whose location/byte_range the batcher relocates when
    # this request is packed into a slab (None = not batchable).
    tensor_entry: Optional[Any] = None


@dataclass
class ReadReq:
    path: str
    consumer: BufferConsumer
    byte_range: Optional[Tuple[int, int]] = None
    # Optional allocator for the read buffer (nbytes -> writable
    # memoryview). Device-targeted consumers hand out pinned memory here so
    # storage reads land directly in DMA-able pages (no bounce copy).
    buf_alloc: Optional[Any] = None


@dataclass
class WriteIO:
    path: str
    buf: BufferType


@dataclass
class ReadIO:
    path: str
    byte_range: Optional[Tuple[int, int]] = None
    buf: Optional[BufferType] = field(default=None)
    buf_alloc: Optional[Any] = None


class StoragePlugin(abc.ABC):
    """A storage backend. All methods are coroutines so network backends can
    interleave many transfers on one event loop; the filesystem backend
    delegates to worker threads (GIL-releasing pwrite/pread)."""

    @abc.abstractmethod
    async def write(self, write_io: WriteIO) -> None:
        ...

    @abc.abstractmethod
    async def read(self, read_io: ReadIO) -> None:
        """Fill ``read_io.buf`` with the (byte-ranged) content of the path."""

    @abc.abstractmethod
    async def delete(self, path: str) -> None:
        ...

    async def delete_dir(self, path: str) -> None:
        raise NotImplementedError

    @abc.abstractmethod
    async def close(self) -> None:
        ...

    async def close_for_loop(self) -> None:
        """Release resources tied to the CURRENT event loop (called when a
        pipeline's loop ends). Backends holding per-loop sessions override
        this; default no-op."""

    # -- sync conveniences (run on a private event loop) --------------------

    def sync_write(self, write_io: WriteIO) -> None:
        from .scheduler import run_coro_sync

        run_coro_sync(self.write(write_io))

    def sync_read(self, read_io: ReadIO) -> None:
        from .scheduler import run_coro_sync

        run_coro_sync(self.read(read_io))

    def sync_close(self) -> None:
        from .scheduler import run_coro_sync

        run_coro_sync(self.close())
