"""Fallback preparer: arbitrary picklable objects via torch.save.

Parity with reference torchsnapshot/io_preparers/object.py:37-95.
"""

from __future__ import annotations

import asyncio
import sys
from typing import Any, List, Tuple

from ..io_types import (
    BufferConsumer,
    BufferStager,
    BufferType,
    ReadReq,
    StageContext,
    WriteReq,
)
from ..manifest import ObjectEntry
from ..serialization import torch_load_from_bytes, torch_save_as_bytes
from .tensor import LoadFuture


class ObjectIOPreparer:
    @staticmethod
    def prepare_write(
        storage_path: str,
        obj: Any,
        replicated: bool = False,
    ) -> Tuple[ObjectEntry, List[WriteReq]]:
        entry = ObjectEntry(
            location=storage_path,
            serializer="torch_save",
            obj_type=type(obj).__name__,
            replicated=replicated,
        )
        return entry, [
            WriteReq(path=storage_path, stager=ObjectBufferStager(obj))
        ]

    @staticmethod
    def prepare_read(entry: ObjectEntry) -> Tuple[List[ReadReq], LoadFuture]:
        fut = LoadFuture()
        return [
            ReadReq(
                path=entry.location,
                consumer=ObjectBufferConsumer(fut=fut),
            )
        ], fut


class ObjectBufferStager(BufferStager):
    def __init__(self, obj: Any) -> None:
        self.obj = obj

    def get_staging_cost_bytes(self) -> int:
        # best-effort estimate; pickling cost is unknowable upfront
        return max(sys.getsizeof(self.obj), 1)

    async def stage_buffer(self, ctx: StageContext) -> BufferType:
        loop = asyncio.get_running_loop()
        return await loop.run_in_executor(
            ctx.executor, torch_save_as_bytes, self.obj
        )


class ObjectBufferConsumer(BufferConsumer):
    def __init__(self, fut: LoadFuture) -> None:
        self.fut = fut

    def get_consuming_cost_bytes(self) -> int:
        return 1

    async def consume_buffer(self, ctx: StageContext, buf: BufferType) -> None:
        loop = asyncio.get_running_loop()
        self.fut.obj = await loop.run_in_executor(
            ctx.executor, torch_load_from_bytes, bytes(buf)
        )
