"""MI355X device staging engine.

Replaces the reference's per-tensor ``tensor.cpu()`` staging
(torchsnapshot/io_preparers/tensor.py:353-355) and its
``GPUBatchedBufferStager`` (torchsnapshot/batcher.py:104-162) with a native
HIP data plane for gfx950:

- a pinned host block pool (D2H lands here; pinned memory is what the SDMA
  engines need to run at full PCIe Gen5 rate, ~55-60 GB/s effective),
- a per-device side HIP stream, so D2H copies and pack-kernel launches
  overlap with whatever the training step has in flight,
- one gather-pack kernel launch per slab: N strided device tensors are
  packed contiguously (256 B aligned each) either into a device slab that
  is then moved by a single SDMA copy, or directly into pinned host memory
  over PCIe (mode knob; both paths are implemented in
  ops/hip/staging_kernels.hip).

Failure policy: if a CUDA/HIP tensor must be staged and the compiled
extension is missing, we raise — a silent eager fallback would invalidate
every benchmark above it. Set TSAMD_DISABLE_HIP_STAGING=1 to explicitly opt
into the torch fallback (debug only).
"""

from __future__ import annotations

import math
import threading
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

import torch

from .. import knobs

ALIGN = 256  # slab alignment per packed tensor

try:
    from torchsnapshot_amd import _csnap  # compiled HIP extension

    HIP_EXT_AVAILABLE = True
except ImportError:  # CPU-only container, or extension not built yet
    _csnap = None
    HIP_EXT_AVAILABLE = False


def _require_ext() -> None:
    if not HIP_EXT_AVAILABLE:
        raise RuntimeError(
            "torchsnapshot_amd's HIP staging extension (_csnap) is not built "
            "but a device tensor needs staging. Build it with "
            "`python -m torchsnapshot_amd.ops.build` (gfx950), or set "
            "TSAMD_DISABLE_HIP_STAGING=1 to explicitly use the slow torch "
            "fallback."
        )


# ---------------------------------------------------------------------------
# tensor layout descriptors
# ---------------------------------------------------------------------------


def collapse_layout(t: torch.Tensor) -> Tuple[List[int], List[int]]:
    """Collapse a tensor's (sizes, strides) by merging ADJACENT LOGICAL dims
    that are jointly contiguous, dropping size-1 dims. Serialization follows
    the logical (row-major over t.shape) element order — the same bytes
    ``t.contiguous()`` would produce — so dims must NOT be reordered by
    stride. Returns element-unit (sizes, strides), innermost last; a fully
    contiguous tensor collapses to ([numel], [1])."""
    sizes = [s for s, st in zip(t.shape, t.stride()) if s != 1]
    strides = [st for s, st in zip(t.shape, t.stride()) if s != 1]
    if not sizes:
        # scalar or all-size-1 tensor: one element
        return [1], [1]
    merged_sizes: List[int] = [sizes[0]]
    merged_strides: List[int] = [strides[0]]
    for s, st in zip(sizes[1:], strides[1:]):
        # logical-adjacent dims merge iff outer stride == inner stride * size
        if merged_strides[-1] == st * s:
            merged_sizes[-1] *= s
            merged_strides[-1] = st
        else:
            merged_sizes.append(s)
            merged_strides.append(st)
    return merged_sizes, merged_strides


@dataclass
class PackItem:
    """One tensor's gather descriptor inside a slab."""

    src_ptr: int          # device base address of element [0,...,0]
    flat_offset: int      # byte offset within the slab
    nbytes: int           # logical payload bytes
    row_bytes: int        # innermost contiguous run in bytes
    outer_sizes: List[int]    # sizes of non-contiguous dims (row-major)
    outer_strides: List[int]  # byte strides of those dims
    vec: int              # safe vector width (16/8/4/2/1)


def build_pack_items(
    tensors: Sequence[torch.Tensor],
) -> Tuple[List[PackItem], List[int], int]:
    """Compute slab layout for a batch: per-tensor descriptors, per-tensor
    slab offsets, and total slab bytes (each tensor 256B-aligned)."""
    items: List[PackItem] = []
    offsets: List[int] = []
    off = 0
    for t in tensors:
        nbytes = t.numel() * t.element_size()
        sizes, strides = collapse_layout(t)
        esz = t.element_size()
        if t.numel() == 0:
            items.append(PackItem(t.data_ptr(), off, 0, 0, [], [], 1))
            offsets.append(off)
            continue
        if strides and strides[-1] == 1:
            row_elems = sizes[-1]
            outer_sizes = sizes[:-1]
            outer_strides_b = [st * esz for st in strides[:-1]]
        else:
            # innermost dim itself strided: rows of one element
            row_elems = 1
            outer_sizes = sizes
            outer_strides_b = [st * esz for st in strides]
        row_bytes = row_elems * esz
        vec = 16
        vec = math.gcd(vec, row_bytes)
        vec = math.gcd(vec, t.data_ptr())
        for st in outer_strides_b:
            vec = math.gcd(vec, st if st else 16)
        items.append(
            PackItem(
                src_ptr=t.data_ptr(),
                flat_offset=off,
                nbytes=nbytes,
                row_bytes=row_bytes,
                outer_sizes=outer_sizes,
                outer_strides=outer_strides_b,
                vec=vec,
            )
        )
        offsets.append(off)
        off += (nbytes + ALIGN - 1) // ALIGN * ALIGN
    return items, offsets, off


def _items_to_flat(items: List[PackItem]) -> List[int]:
    """Serialize descriptors for the extension: fixed-width int rows."""
    MAXD = 6
    flat: List[int] = []
    for it in items:
        if len(it.outer_sizes) > MAXD:
            raise ValueError(
                f"tensor layout too complex to pack ({len(it.outer_sizes)} "
                "outer dims after collapsing; max 6)"
            )
        row = [
            it.src_ptr,
            it.flat_offset,
            it.nbytes,
            it.row_bytes,
            it.vec,
            len(it.outer_sizes),
        ]
        sizes = list(it.outer_sizes) + [1] * (MAXD - len(it.outer_sizes))
        strides = list(it.outer_strides) + [0] * (MAXD - len(it.outer_strides))
        row += sizes + strides
        flat += row
    return flat


# ---------------------------------------------------------------------------
# pinned host pool
# ---------------------------------------------------------------------------


class PinnedBlock:
    def __init__(self, tensor: torch.Tensor, pooled: bool) -> None:
        self.tensor = tensor  # 1-D uint8, pin_memory=True
        self.pooled = pooled

    @property
    def nbytes(self) -> int:
        return self.tensor.numel()


class PinnedPool:
    """Fixed pool of pinned host blocks; oversized requests get a one-off
    pinned allocation. acquire() blocks until a block frees up, which
    naturally backpressures staging against storage-write drain."""

    def __init__(
        self, block_size: Optional[int] = None, block_count: Optional[int] = None
    ) -> None:
        self.block_size = block_size or knobs.get_pinned_block_size_bytes()
        self.max_blocks = block_count or knobs.get_pinned_block_count()
        self._free: List[torch.Tensor] = []
        self._allocated = 0
        self._cond = threading.Condition()

    def acquire(self, nbytes: int) -> PinnedBlock:
        if nbytes > self.block_size:
            return PinnedBlock(
                torch.empty(nbytes, dtype=torch.uint8, pin_memory=True),
                pooled=False,
            )
        with self._cond:
            while True:
                if self._free:
                    return PinnedBlock(self._free.pop(), pooled=True)
                if self._allocated < self.max_blocks:
                    self._allocated += 1
                    break
                self._cond.wait()
        return PinnedBlock(
            torch.empty(self.block_size, dtype=torch.uint8, pin_memory=True),
            pooled=True,
        )

    def release(self, block: PinnedBlock) -> None:
        if not block.pooled:
            return  # one-off allocation; freed by GC
        with self._cond:
            self._free.append(block.tensor)
            self._cond.notify()

    def prealloc_one(self) -> bool:
        """Allocate one new block straight into the free list (never takes
        from it), so warming can proceed concurrently with real staging
        without hoarding blocks."""
        with self._cond:
            if self._allocated >= self.max_blocks:
                return False
            self._allocated += 1
        tensor = torch.empty(self.block_size, dtype=torch.uint8, pin_memory=True)
        with self._cond:
            self._free.append(tensor)
            self._cond.notify()
        return True


_pool_lock = threading.Lock()
_pool: Optional[PinnedPool] = None


def get_pinned_pool() -> PinnedPool:
    global _pool
    with _pool_lock:
        if _pool is None:
            _pool = PinnedPool()
        return _pool


_warm_stop = threading.Event()
_warm_thread: Optional[threading.Thread] = None


def _join_warm_thread() -> None:
    # a pinned allocation in flight during interpreter teardown aborts the
    # process (HIP resources die under the daemon thread); stop between
    # blocks and join before exit
    _warm_stop.set()
    t = _warm_thread
    if t is not None and t.is_alive():
        t.join(timeout=30)


def warm_pinned_pool(nbytes: Optional[int] = None, background: bool = True) -> None:
    """Pre-allocate (and so pre-register) pinned blocks. Pinning is a
    one-time ~GB/s kernel-side cost; warming it off the critical path
    keeps the FIRST checkpoint as fast as the rest (measured 14.4 s ->
    0.35 s first-async-take stall for an 8 GB model). Called automatically
    (in the background) when a StagingEngine is first created."""
    global _warm_thread
    pool = get_pinned_pool()

    def work() -> None:
        import os

        if nbytes is not None:
            target = nbytes
        else:
            # default: don't pin the whole pool for workloads that may
            # never need it — 8 GB covers most first checkpoints
            target = min(
                pool.block_size * pool.max_blocks,
                int(float(os.environ.get("TSAMD_POOL_WARM_BYTES", 8 * 1024**3))),
            )
        allocated = 0
        while (
            not _warm_stop.is_set()
            and allocated < target
            and pool.prealloc_one()
        ):
            allocated += pool.block_size

    if background:
        import atexit

        with _pool_lock:
            if _warm_thread is not None and _warm_thread.is_alive():
                return
            _warm_thread = threading.Thread(
                target=work, name="tsamd-pool-warm", daemon=True
            )
            atexit.register(_join_warm_thread)
            _warm_thread.start()
    else:
        work()


# ---------------------------------------------------------------------------
# staging engine
# ---------------------------------------------------------------------------


@dataclass
class StagedBatch:
    """Handle for one in-flight D2H staging operation."""

    pinned: PinnedBlock
    offsets: List[int]
    nbytes_list: List[int]
    total_bytes: int
    _handle: Optional[int] = None          # _csnap op handle
    _device_slab: Optional[torch.Tensor] = None  # keep alive until done
    # per-tensor psum64 values (device-computed), populated at wait() when
    # checksumming was requested
    checksums: Optional[List[int]] = None
    _hash_tensor: Optional[torch.Tensor] = None
    # source tensors referenced until the async gather/copy completes, so
    # callers may drop theirs right after stage()
    _sources: Optional[Sequence[torch.Tensor]] = None
    _done: bool = False

    def wait(self) -> None:
        """Block until the D2H copy has landed in pinned memory."""
        if self._done:
            return
        if self._handle is not None:
            _csnap.wait(self._handle)
            self._handle = None
        if self._hash_tensor is not None:
            self.checksums = [int(v) for v in self._hash_tensor.cpu().tolist()]
            self._hash_tensor = None
        self._device_slab = None
        self._sources = None
        self._done = True

    def memoryview_of(self, index: int) -> memoryview:
        assert self._done, "wait() before reading staged buffers"
        off = self.offsets[index]
        n = self.nbytes_list[index]
        if n == 0:
            return memoryview(b"")
        return memoryview(self.pinned.tensor.numpy())[off : off + n]

    def slab_memoryview(self) -> memoryview:
        assert self._done
        return memoryview(self.pinned.tensor.numpy())[: self.total_bytes]

    def release(self) -> None:
        get_pinned_pool().release(self.pinned)


class StagingEngine:
    """Per-device staging front end. Thread-safe; all GPU work goes to the
    extension's side stream for the device."""

    def __init__(self, device: torch.device) -> None:
        self.device = device
        self._use_ext = HIP_EXT_AVAILABLE and not knobs.is_hip_staging_disabled()
        if not self._use_ext and not knobs.is_hip_staging_disabled():
            _require_ext()
        # hide the one-time pinned-registration cost behind whatever runs
        # before the first checkpoint
        warm_pinned_pool()

    def stage(
        self,
        tensors: Sequence[torch.Tensor],
        compute_checksums: bool = False,
    ) -> StagedBatch:
        """Start async D2H staging of device tensors into one pinned slab.

        With ``compute_checksums``, the gather kernel also accumulates a
        psum64 checksum per tensor at no measurable cost (the kernel is
        PCIe-bound); padding gaps are zeroed so the sum of the per-tensor
        values is the checksum of the whole written file.

        The caller must ensure the tensors' producing stream is
        torch.cuda.current_stream() of this thread (true for checkpointing:
        tensors are live parameters/opt states, already materialized)."""
        # the pack kernel indexes within-tensor bytes as u32: materialize
        # non-contiguous tensors over 2 GiB first (chunking splits along
        # dim 0 upstream, so this is rare). The stride>=0 check is
        # defensive only — torch itself forbids negative strides.
        tensors = [
            t
            if (
                t.is_contiguous()
                or (
                    t.numel() * t.element_size() < 2**31
                    and all(s >= 0 for s in t.stride())
                )
            )
            else t.contiguous()
            for t in tensors
        ]
        items, offsets, total = build_pack_items(tensors)
        nbytes_list = [it.nbytes for it in items]
        pinned = get_pinned_pool().acquire(max(total, 1))
        batch = StagedBatch(
            pinned=pinned,
            offsets=offsets,
            nbytes_list=nbytes_list,
            total_bytes=total,
        )
        if total == 0:
            batch._done = True
            return batch
        batch._sources = tensors
        try:
            if self._use_ext:
                self._stage_ext(
                    tensors, items, batch, total,
                    compute_checksums=compute_checksums,
                )
            else:
                self._stage_torch_fallback(tensors, items, batch)
        except BaseException:
            # don't leak the pool slot on launch failure
            get_pinned_pool().release(pinned)
            raise
        return batch

    # -- native path --------------------------------------------------------

    def _stage_ext(
        self,
        tensors: Sequence[torch.Tensor],
        items: List[PackItem],
        batch: StagedBatch,
        total: int,
        compute_checksums: bool = False,
    ) -> None:
        dev_index = self.device.index or 0
        with torch.cuda.device(dev_index):
            cur_stream = torch.cuda.current_stream().cuda_stream
            single_contig = (
                len(items) == 1
                and not items[0].outer_sizes
                and not compute_checksums
            )
            if single_contig:
                # one SDMA copy, no kernel
                handle = _csnap.d2h_copy(
                    items[0].src_ptr,
                    batch.pinned.tensor.data_ptr(),
                    items[0].nbytes,
                    cur_stream,
                    dev_index,
                )
            else:
                mode = _pack_mode()
                slab = None
                slab_ptr = 0
                hash_ptr = 0
                if compute_checksums:
                    # padding must read as zeros so per-tensor checksums
                    # sum to the file checksum
                    batch._hash_tensor = torch.zeros(
                        len(items), dtype=torch.uint64, device=self.device
                    )
                    hash_ptr = batch._hash_tensor.data_ptr()
                if mode == "slab":
                    slab = (
                        torch.zeros(total, dtype=torch.uint8, device=self.device)
                        if compute_checksums
                        else torch.empty(
                            total, dtype=torch.uint8, device=self.device
                        )
                    )
                    slab_ptr = slab.data_ptr()
                elif compute_checksums:
                    # direct mode: kernel writes only payload bytes; zero
                    # the pinned gap regions host-side (tiny)
                    pin_np = batch.pinned.tensor.numpy()
                    prev_end = 0
                    for it in items:
                        if it.flat_offset > prev_end:
                            pin_np[prev_end : it.flat_offset] = 0
                        prev_end = it.flat_offset + it.nbytes
                    if total > prev_end:
                        pin_np[prev_end:total] = 0
                handle = _csnap.pack_d2h(
                    _items_to_flat(items),
                    len(items),
                    slab_ptr,
                    batch.pinned.tensor.data_ptr(),
                    total,
                    cur_stream,
                    dev_index,
                    hash_ptr,
                )
                batch._device_slab = slab
            batch._handle = handle

    # -- debug fallback -----------------------------------------------------

    def _stage_torch_fallback(
        self,
        tensors: Sequence[torch.Tensor],
        items: List[PackItem],
        batch: StagedBatch,
    ) -> None:
        for t, it in zip(tensors, items):
            if it.nbytes == 0:
                continue
            flat = t.contiguous().reshape(-1).view(torch.uint8)
            dst = batch.pinned.tensor[it.flat_offset : it.flat_offset + it.nbytes]
            dst.copy_(flat, non_blocking=True)
        torch.cuda.synchronize(self.device)
        batch._done = True


def _pack_mode() -> str:
    # "direct": the gather kernel writes straight into pinned host memory
    # over PCIe (measured ~50 GB/s on MI355X, and +20% on the full
    # checkpoint bench vs the device-slab + SDMA hop, which also costs a
    # transient slab allocation per batch). "slab" keeps CU time minimal
    # (kernel runs at HBM speed, SDMA does the PCIe leg) for overlap with
    # heavy training compute.
    import os

    mode = os.environ.get("TSAMD_STAGE_MODE", "direct")
    if mode not in ("slab", "direct"):
        raise ValueError(f"TSAMD_STAGE_MODE must be slab|direct, got {mode}")
    return mode


def copy_buffer_via_pinned(
    buf, dtype: torch.dtype, shape: Sequence[int], device: torch.device
) -> torch.Tensor:
    """Restore fast path: host buffer -> pinned bounce -> SDMA H2D.

    Returns a contiguous device tensor with the buffer's content. The
    pageable-copy alternative (torch.frombuffer(...).to(cuda)) measures
    ~3x slower on MI355X."""
    mv = memoryview(buf)
    nbytes = mv.nbytes
    pool = get_pinned_pool()
    block = pool.acquire(max(nbytes, 1))
    try:
        src = torch.frombuffer(mv, dtype=torch.uint8)
        block.tensor[:nbytes].copy_(src)  # releases the GIL
        dev_u8 = block.tensor[:nbytes].to(device, non_blocking=False)
    finally:
        pool.release(block)
    if dtype == torch.uint8:
        return dev_u8.reshape(tuple(shape))
    return dev_u8.view(dtype).reshape(tuple(shape))


def device_psum64(dev_u8: torch.Tensor, file_byte_off: int = 0) -> int:
    """psum64 of a contiguous uint8 device tensor, computed on-device at
    HBM speed (restore-verification path; ~free vs the H2D copy that
    produced the bytes). ``file_byte_off`` is the byte offset of the
    buffer's first byte within its payload file (8-aligned)."""
    _require_ext()
    assert dev_u8.dtype == torch.uint8 and dev_u8.is_contiguous()
    assert dev_u8.device.type == "cuda"
    if dev_u8.numel() == 0:
        return 0
    return int(
        _csnap.psum64_dev(
            dev_u8.data_ptr(),
            dev_u8.numel(),
            file_byte_off,
            torch.cuda.current_stream(dev_u8.device).cuda_stream,
            dev_u8.device.index,
        )
    )


def verify_device_psum(
    dev_u8: torch.Tensor, expected: "tuple[int, int]", what: str
) -> None:
    """Check a device buffer against (expected_psum_value, word_base);
    raises like the CPU verifier on mismatch."""
    value, word_base = expected
    got = device_psum64(dev_u8, word_base * 8)
    if got != value:
        raise RuntimeError(
            f"checksum mismatch for '{what}': snapshot recorded "
            f"psum64:{value:016x}, device read back psum64:{got:016x} — "
            "the file is corrupted or was modified after the snapshot "
            "was committed"
        )


_engines: Dict[int, StagingEngine] = {}
_engines_lock = threading.Lock()


def get_staging_engine(device: torch.device) -> StagingEngine:
    idx = device.index if device.index is not None else torch.cuda.current_device()
    with _engines_lock:
        if idx not in _engines:
            _engines[idx] = StagingEngine(torch.device("cuda", idx))
        return _engines[idx]
