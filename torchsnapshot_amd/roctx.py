"""Optional roctx markers: stage/write/read phases show up as named
ranges in rocprofv3 timelines (--sys-trace marker domain). No-ops when
libroctx64 is unavailable (CPU containers)."""

from __future__ import annotations

import contextlib
import ctypes
from typing import Iterator, Optional

_lib: Optional[ctypes.CDLL] = None
_tried = False


def _roctx() -> Optional[ctypes.CDLL]:
    global _lib, _tried
    if not _tried:
        _tried = True
        for name in ("libroctx64.so", "/opt/rocm/lib/libroctx64.so"):
            try:
                _lib = ctypes.CDLL(name)
                _lib.roctxRangePushA.argtypes = [ctypes.c_char_p]
                _lib.roctxRangePop.argtypes = []
                break
            except OSError:
                continue
    return _lib


@contextlib.contextmanager
def roctx_range(name: str) -> Iterator[None]:
    lib = _roctx()
    if lib is None:
        yield
        return
    lib.roctxRangePushA(name.encode())
    try:
        yield
    finally:
        lib.roctxRangePop()
