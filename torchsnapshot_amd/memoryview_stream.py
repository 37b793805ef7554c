"""A read-only file-like object over a memoryview.

Lets zero-copy tensor buffers be streamed to HTTP clients (S3 uploads)
without materializing bytes (parity with reference
torchsnapshot/memoryview_stream.py:14-87).
"""

from __future__ import annotations

import io
from typing import Optional


class MemoryviewStream(io.RawIOBase):
    def __init__(self, mv: memoryview) -> None:
        super().__init__()
        self._mv = mv.cast("B") if mv.format != "B" else mv
        self._pos = 0

    def readable(self) -> bool:
        return True

    def seekable(self) -> bool:
        return True

    def seek(self, pos: int, whence: int = io.SEEK_SET) -> int:
        if self.closed:
            raise ValueError("I/O operation on closed stream")
        if whence == io.SEEK_SET:
            new_pos = pos
        elif whence == io.SEEK_CUR:
            new_pos = self._pos + pos
        elif whence == io.SEEK_END:
            new_pos = len(self._mv) + pos
        else:
            raise ValueError(f"invalid whence: {whence}")
        if new_pos < 0:
            raise ValueError(f"negative seek position: {new_pos}")
        self._pos = new_pos
        return self._pos

    def tell(self) -> int:
        if self.closed:
            raise ValueError("I/O operation on closed stream")
        return self._pos

    def read(self, size: Optional[int] = -1) -> bytes:
        if self.closed:
            raise ValueError("I/O operation on closed stream")
        if size is None or size < 0:
            end = len(self._mv)
        else:
            end = min(self._pos + size, len(self._mv))
        data = bytes(self._mv[self._pos : end])
        self._pos = end
        return data

    def readinto(self, b) -> int:
        data = self.read(len(b))
        n = len(data)
        b[:n] = data
        return n

    def close(self) -> None:
        super().close()
