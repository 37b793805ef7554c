"""Tensor I/O preparer: the core of the data path.

Write: a device tensor is staged through the HIP engine (gather-pack kernel
+ SDMA D2H into pinned memory, ops/staging.py); a CPU tensor is exposed
zero-copy (cloned only when required for safety). Read: buffer bytes are
wrapped zero-copy and scattered into the destination tensor in-place,
with optional tiled reads that bound peak host memory for huge tensors.

Parity with reference torchsnapshot/io_preparers/tensor.py (staging rules
:240-307, tiled reads :128-181, quantized-aware copy :385-409) with an
MI355X-native staging mechanism.
"""

from __future__ import annotations

import asyncio
from typing import Any, List, Optional, Tuple

import torch

from ..io_types import (
    BufferConsumer,
    BufferStager,
    BufferType,
    ReadReq,
    StageContext,
    WriteReq,
)
from ..manifest import TensorEntry
from ..serialization import (
    SERIALIZER_BUFFER,
    SERIALIZER_QTENSOR,
    SERIALIZER_TORCH_SAVE,
    dtype_to_str,
    pick_serializer,
    qtensor_as_bytes,
    qtensor_from_bytes,
    str_to_dtype,
    tensor_as_memoryview,
    tensor_from_memoryview,
    torch_load_from_bytes,
    torch_save_as_bytes,
)


class LoadFuture:
    """Resolved once the read pipeline completes; .obj is the loaded value."""

    __slots__ = ["obj"]

    def __init__(self, obj: Any = None) -> None:
        self.obj = obj


def tensor_copy(dst: torch.Tensor, src: torch.Tensor) -> None:
    """Copy src into dst in-place, handling dtype/device casts and
    quantization-scheme mismatches."""
    dst = dst.detach()
    if src.is_quantized and not dst.is_quantized:
        src = src.dequantize()
    if dst.is_quantized and not src.is_quantized:
        raise ValueError(
            "cannot load a non-quantized payload into a quantized tensor "
            f"(dst dtype {dst.dtype})"
        )
    if (
        dst.is_quantized
        and src.is_quantized
        and dst.qscheme() != src.qscheme()
    ):
        raise ValueError(
            f"quantization scheme mismatch: dst {dst.qscheme()} vs src "
            f"{src.qscheme()}"
        )
    dst.copy_(src)


class TensorIOPreparer:
    @staticmethod
    def prepare_write(
        storage_path: str,
        tensor: torch.Tensor,
        replicated: bool = False,
        is_async_snapshot: bool = False,
    ) -> Tuple[TensorEntry, List[WriteReq]]:
        serializer = pick_serializer(tensor)
        entry = TensorEntry(
            location=storage_path,
            serializer=serializer,
            dtype=dtype_to_str(tensor.dtype),
            shape=list(tensor.shape),
            replicated=replicated,
        )
        stager = TensorBufferStager(
            tensor=tensor,
            serializer=serializer,
            is_async_snapshot=is_async_snapshot,
        )
        return entry, [
            WriteReq(path=storage_path, stager=stager, tensor_entry=entry)
        ]

    @staticmethod
    def prepare_read(
        entry: TensorEntry,
        tensor_out: Optional[torch.Tensor] = None,
        buffer_size_limit_bytes: Optional[int] = None,
    ) -> Tuple[List[ReadReq], LoadFuture]:
        nbytes = entry.nbytes_estimate()
        if (
            buffer_size_limit_bytes is not None
            and entry.serializer == SERIALIZER_BUFFER
            and entry.byte_range is None
            and nbytes > buffer_size_limit_bytes
        ):
            return TensorIOPreparer._prepare_read_tiled(
                entry, tensor_out, buffer_size_limit_bytes
            )
        fut = LoadFuture(tensor_out)
        consumer = TensorBufferConsumer(entry=entry, tensor_out=tensor_out, fut=fut)
        byte_range = tuple(entry.byte_range) if entry.byte_range else None
        # device targets read straight into pinned memory (if the request
        # ends up merged into a batched span, the span read ignores this
        # and the consumer falls back to the bounce path)
        buf_alloc = (
            consumer.alloc_pinned_buffer
            if (
                tensor_out is not None
                and tensor_out.device.type == "cuda"
                and entry.serializer == SERIALIZER_BUFFER
            )
            else None
        )
        return [
            ReadReq(
                path=entry.location,
                consumer=consumer,
                byte_range=byte_range,
                buf_alloc=buf_alloc,
            )
        ], fut

    @staticmethod
    def _prepare_read_tiled(
        entry: TensorEntry,
        tensor_out: Optional[torch.Tensor],
        limit: int,
    ) -> Tuple[List[ReadReq], LoadFuture]:
        """Split one big buffer-serialized tensor into byte-ranged reads so
        peak host memory stays under ``limit`` (random-access API)."""
        dtype = str_to_dtype(entry.dtype)
        nbytes = entry.nbytes_estimate()
        elem = max(dtype.itemsize, 1)
        # tiles write straight into the output when it is contiguous with a
        # matching dtype — on ANY device (CUDA tiles go pinned -> SDMA H2D,
        # so peak host memory stays at the budget, never the tensor size)
        direct = (
            tensor_out is not None
            and tensor_out.is_contiguous()
            and tensor_out.dtype == dtype
        )
        staging = (
            tensor_out
            if direct
            else torch.empty(entry.shape, dtype=dtype)
        )
        fut = LoadFuture(tensor_out if tensor_out is not None else staging)
        flat_u8 = staging.reshape(-1).view(torch.uint8) if nbytes else None
        tile = max(limit - limit % elem, elem)
        reqs: List[ReadReq] = []
        state = _TiledReadState(
            remaining=(nbytes + tile - 1) // tile if nbytes else 0,
            staging=staging,
            tensor_out=None if direct else tensor_out,
        )
        for start in range(0, nbytes, tile):
            end = min(start + tile, nbytes)
            consumer = _TensorTileConsumer(
                dst_flat_u8=flat_u8, start=start, end=end, state=state
            )
            reqs.append(
                ReadReq(
                    path=entry.location,
                    byte_range=(start, end),
                    consumer=consumer,
                    buf_alloc=(
                        consumer.alloc_pinned_buffer
                        if staging.device.type == "cuda"
                        else None
                    ),
                )
            )
        if not reqs:  # empty tensor
            if tensor_out is not None and not direct:
                tensor_copy(tensor_out, staging)
        return reqs, fut


class _TiledReadState:
    def __init__(
        self,
        remaining: int,
        staging: torch.Tensor,
        tensor_out: Optional[torch.Tensor],
    ) -> None:
        self.remaining = remaining
        self.staging = staging
        self.tensor_out = tensor_out
        self.lock = __import__("threading").Lock()


class _TensorTileConsumer(BufferConsumer):
    def __init__(
        self,
        dst_flat_u8: torch.Tensor,
        start: int,
        end: int,
        state: _TiledReadState,
    ) -> None:
        self.dst_flat_u8 = dst_flat_u8
        self.start = start
        self.end = end
        self.state = state
        self._pinned_block = None

    def alloc_pinned_buffer(self, nbytes: int) -> memoryview:
        from ..ops.staging import get_pinned_pool

        self._pinned_block = get_pinned_pool().acquire(max(nbytes, 1))
        return memoryview(self._pinned_block.tensor.numpy())[:nbytes]

    def close(self) -> None:
        if self._pinned_block is not None:
            from ..ops.staging import get_pinned_pool

            get_pinned_pool().release(self._pinned_block)
            self._pinned_block = None

    def get_consuming_cost_bytes(self) -> int:
        return self.end - self.start

    async def consume_buffer(self, ctx: StageContext, buf: BufferType) -> None:
        def work() -> None:
            n = self.end - self.start
            if self._pinned_block is not None:
                src = self._pinned_block.tensor[:n]  # pinned: async-capable
            else:
                src = torch.frombuffer(buf, dtype=torch.uint8)
            self.dst_flat_u8[self.start : self.end].copy_(src)
            with self.state.lock:
                self.state.remaining -= 1
                last = self.state.remaining == 0
            if last and self.state.tensor_out is not None:
                tensor_copy(self.state.tensor_out, self.state.staging)

        await asyncio.get_running_loop().run_in_executor(ctx.executor, work)


class TensorBufferStager(BufferStager):
    def __init__(
        self,
        tensor: torch.Tensor,
        serializer: str,
        is_async_snapshot: bool = False,
    ) -> None:
        self.tensor = tensor
        self.serializer = serializer
        self.is_async_snapshot = is_async_snapshot
        self._staged_batch = None  # StagedBatch for device tensors
        # device staging computes this for free when TSAMD_CHECKSUM=1
        self.precomputed_checksum = None

    def get_staging_cost_bytes(self) -> int:
        nbytes = self.tensor.numel() * self.tensor.element_size()
        if self.serializer in (SERIALIZER_TORCH_SAVE, SERIALIZER_QTENSOR):
            # these serializers materialize a second copy while packing
            return 2 * nbytes
        return nbytes

    async def stage_buffer(self, ctx: StageContext) -> BufferType:
        loop = asyncio.get_running_loop()
        if self.tensor.device.type == "cuda":
            return await loop.run_in_executor(ctx.executor, self._stage_device)
        return await loop.run_in_executor(ctx.executor, self._stage_cpu, ctx)

    def _stage_device(self) -> BufferType:
        from ..ops.staging import get_staging_engine
        from ..uvm_tensor import is_uvm_tensor, uvm_to_cpu

        t = self.tensor.detach()
        if t.is_quantized or self.serializer != SERIALIZER_BUFFER:
            # rare path: bring to host with torch, then pack
            cpu = t.cpu()
            if self.serializer == SERIALIZER_QTENSOR:
                return qtensor_as_bytes(cpu)
            return torch_save_as_bytes(cpu)
        if is_uvm_tensor(t):
            # managed memory is CPU-addressable: serialize zero-copy, no
            # D2H needed (clone when async: training may mutate the pages)
            cpu = uvm_to_cpu(t)
            if self.is_async_snapshot:
                cpu = cpu.clone()
            return tensor_as_memoryview(cpu.contiguous())
        from .. import integrity

        engine = get_staging_engine(t.device)
        ck = integrity.checksumming_enabled()
        batch = engine.stage([t], compute_checksums=ck)
        batch.wait()  # blocks in executor thread; GIL released inside HIP
        if ck and batch.checksums is not None:
            self.precomputed_checksum = "psum64:" + format(
                batch.checksums[0], "016x"
            )
        self._staged_batch = batch
        return batch.memoryview_of(0)

    def _stage_cpu(self, ctx: StageContext) -> BufferType:
        t = self.tensor.detach()
        if self.serializer == SERIALIZER_QTENSOR:
            return qtensor_as_bytes(t)
        if self.serializer == SERIALIZER_TORCH_SAVE:
            return torch_save_as_bytes(t)
        if self._should_copy_cpu_tensor(t, ctx):
            t = t.contiguous().clone() if not t.is_contiguous() else t.clone()
        elif not t.is_contiguous():
            t = t.contiguous()
        return tensor_as_memoryview(t)

    def _should_copy_cpu_tensor(self, t: torch.Tensor, ctx: StageContext) -> bool:
        # Async snapshots: training resumes (and may mutate the tensor) as
        # soon as staging completes, so the bytes handed to storage must be
        # a private copy.
        if self.is_async_snapshot:
            return True
        # A view over a larger storage: serializing zero-copy would leak
        # unrelated bytes lifetimes; cheaper to copy the logical content.
        if t.numel() * t.element_size() != t.untyped_storage().nbytes():
            return True
        return False

    def release_buffer(self) -> None:
        if self._staged_batch is not None:
            self._staged_batch.release()
            self._staged_batch = None
        # drop the source reference: for shadow-cloned async saves this
        # frees the device clone as soon as its write completes (any
        # zero-copy host buffer stays alive through the memoryview)
        self.tensor = None  # type: ignore[assignment]


class TensorBufferConsumer(BufferConsumer):
    def __init__(
        self,
        entry: TensorEntry,
        tensor_out: Optional[torch.Tensor],
        fut: LoadFuture,
    ) -> None:
        self.entry = entry
        self.tensor_out = tensor_out
        self.fut = fut
        self._pinned_block = None
        self._pinned_nbytes = 0
        # (expected psum64 value, word_base) set by the read scheduler
        # when this consumer verifies on device (will_verify_on_device)
        self.expected_psum = None

    def alloc_pinned_buffer(self, nbytes: int) -> memoryview:
        """ReadReq.buf_alloc hook: the storage layer reads directly into
        pinned memory, so H2D needs no bounce copy."""
        from ..ops.staging import get_pinned_pool

        self._pinned_block = get_pinned_pool().acquire(max(nbytes, 1))
        self._pinned_nbytes = nbytes
        return memoryview(self._pinned_block.tensor.numpy())[:nbytes]

    def close(self) -> None:
        if self._pinned_block is not None:
            from ..ops.staging import get_pinned_pool

            get_pinned_pool().release(self._pinned_block)
            self._pinned_block = None

    def get_consuming_cost_bytes(self) -> int:
        nbytes = self.entry.nbytes_estimate()
        if self.entry.serializer == SERIALIZER_TORCH_SAVE:
            return 2 * nbytes
        return nbytes

    def device_span_target(self):
        # non-None when this consumer can take its bytes as a slice of a
        # device-resident uint8 span (batched-span restore fast path)
        if (
            self.tensor_out is not None
            and self.tensor_out.device.type == "cuda"
            and self.entry.serializer == SERIALIZER_BUFFER
        ):
            return self.tensor_out.device
        return None

    def will_verify_on_device(self) -> bool:
        """True when consume_buffer will take the pinned->H2D device path
        and can therefore checksum the bytes on-device at HBM speed
        instead of CPU-hashing them in the read pipeline."""
        from ..ops.staging import HIP_EXT_AVAILABLE

        return (
            HIP_EXT_AVAILABLE
            and self._pinned_block is not None
            and self.tensor_out is not None
            and self.tensor_out.device.type == "cuda"
            and self.entry.serializer == SERIALIZER_BUFFER
        )

    def consume_from_device_u8(self, dev_u8: torch.Tensor) -> None:
        dtype = str_to_dtype(self.entry.dtype)
        loaded = (
            dev_u8.view(dtype).reshape(tuple(self.entry.shape))
            if dtype != torch.uint8
            else dev_u8.reshape(tuple(self.entry.shape))
        )
        tensor_copy(self.tensor_out, loaded)
        self.fut.obj = self.tensor_out

    async def consume_buffer(self, ctx: StageContext, buf: BufferType) -> None:
        def work() -> None:
            dtype = (
                str_to_dtype(self.entry.dtype)
                if self.entry.serializer == SERIALIZER_BUFFER
                else None
            )
            if self.entry.serializer == SERIALIZER_QTENSOR:
                loaded = qtensor_from_bytes(buf)
            elif self.entry.serializer == SERIALIZER_TORCH_SAVE:
                loaded = torch_load_from_bytes(bytes(buf))
            elif (
                self._pinned_block is not None
                and self.tensor_out is not None
                and self.tensor_out.device.type == "cuda"
            ):
                # buffer already lives in pinned memory: straight SDMA H2D
                n = self._pinned_nbytes
                dev_u8 = self._pinned_block.tensor[:n].to(
                    self.tensor_out.device, non_blocking=False
                )
                if self.expected_psum is not None:
                    from ..ops.staging import verify_device_psum

                    verify_device_psum(
                        dev_u8, self.expected_psum, self.entry.location
                    )
                loaded = (
                    dev_u8.view(dtype).reshape(tuple(self.entry.shape))
                    if dtype != torch.uint8
                    else dev_u8.reshape(tuple(self.entry.shape))
                )
            elif (
                self.tensor_out is not None
                and self.tensor_out.device.type == "cuda"
            ):
                # batched-span slice: pinned bounce + SDMA H2D
                from ..ops.staging import copy_buffer_via_pinned

                loaded = copy_buffer_via_pinned(
                    buf,
                    dtype=dtype,
                    shape=tuple(self.entry.shape),
                    device=self.tensor_out.device,
                )
            else:
                loaded = tensor_from_memoryview(
                    memoryview(buf), dtype=dtype, shape=tuple(self.entry.shape)
                )
            if self.tensor_out is not None:
                tensor_copy(self.tensor_out, loaded)
                self.fut.obj = self.tensor_out
            else:
                self.fut.obj = loaded

        await asyncio.get_running_loop().run_in_executor(ctx.executor, work)
