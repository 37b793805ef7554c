"""Unified (managed) memory tensors.

Embedding tables larger than one GPU's HBM can live in hipMallocManaged
memory that both the CPU and the GPU address. The staging path treats a
UVM tensor's bytes as host-readable and serializes them zero-copy,
skipping the D2H copy entirely (the reference gates this on fbgemm_gpu
ops, torchsnapshot/uvm_tensor.py:15-48; here it is native _csnap).

``new_managed_tensor`` returns a CPU-viewed torch tensor whose storage is
managed memory: CPU code (and this library's serialization) reads it
in place; HIP kernels can address the same pages through the raw pointer.
"""

from __future__ import annotations

import ctypes
from typing import Sequence

import numpy as np
import torch


def _csnap_or_none():
    try:
        from torchsnapshot_amd import _csnap

        return _csnap
    except ImportError:
        return None


def is_uvm_tensor(t: torch.Tensor) -> bool:
    """True if the tensor's storage is hipMallocManaged memory."""
    ext = _csnap_or_none()
    if ext is None:
        return False
    try:
        return bool(ext.is_managed_ptr(t.data_ptr()))
    except Exception:
        return False


def uvm_to_cpu(t: torch.Tensor) -> torch.Tensor:
    """A zero-copy CPU view of a UVM tensor's memory. Falls back to a
    (page-migrating) .cpu() copy for non-managed device tensors."""
    if not t.is_cuda:
        return t
    if not is_uvm_tensor(t):
        return t.cpu()
    if not t.is_contiguous():
        return t.cpu()
    nbytes = t.numel() * t.element_size()
    raw = (ctypes.c_uint8 * nbytes).from_address(t.data_ptr())
    u8 = torch.from_numpy(np.ctypeslib.as_array(raw))
    view = u8.view(t.dtype).reshape(t.shape)
    # keep the source tensor alive as long as the view exists
    view._tsamd_uvm_owner = t  # type: ignore[attr-defined]
    return view


def new_managed_tensor(
    shape: Sequence[int], dtype: torch.dtype = torch.float32, device: int = 0
) -> torch.Tensor:
    """Allocate a managed (UVM) tensor via hipMallocManaged, advised to
    prefer host residency (the embedding-table pattern). Returns a CPU-view
    tensor; the same pages are GPU-addressable via data_ptr()."""
    ext = _csnap_or_none()
    if ext is None:
        raise RuntimeError(
            "managed-tensor allocation needs the _csnap extension "
            "(python -m torchsnapshot_amd.ops.build)"
        )
    numel = 1
    for s in shape:
        numel *= int(s)
    nbytes = max(numel * dtype.itemsize, 1)
    ptr = ext.managed_alloc(nbytes, device)
    ext.managed_advise_preferred_cpu(ptr, nbytes)
    raw = (ctypes.c_uint8 * nbytes).from_address(ptr)
    u8 = torch.from_numpy(np.ctypeslib.as_array(raw))
    t = u8.view(dtype).reshape(tuple(shape))

    # free the managed allocation when the tensor goes away
    import weakref

    weakref.finalize(t, ext.managed_free, ptr)
    return t


def prefetch_to_device(t: torch.Tensor, device: int) -> None:
    """Migrate a managed tensor's pages toward a device (-1 = CPU)."""
    ext = _csnap_or_none()
    if ext is None or not is_uvm_tensor(t):
        return
    ext.managed_prefetch(t.data_ptr(), t.numel() * t.element_size(), device)
