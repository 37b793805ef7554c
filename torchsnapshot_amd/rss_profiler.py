"""RSS profiling: sample resident-set-size deltas on a background thread.

Used by benchmarks to demonstrate the memory-budgeted pipeline keeps peak
host memory bounded (parity with reference torchsnapshot/rss_profiler.py).
"""

from __future__ import annotations

import contextlib
import threading
import time
from typing import Deque, Iterator

import psutil

_MB = 1024 * 1024


@contextlib.contextmanager
def measure_rss_deltas(rss_deltas: Deque[int], interval_s: float = 0.1) -> Iterator[None]:
    """Appends (rss - baseline) samples, in bytes, to ``rss_deltas`` every
    ``interval_s`` while the context is active."""
    proc = psutil.Process()
    baseline = proc.memory_info().rss
    stop = threading.Event()

    def sampler() -> None:
        while not stop.is_set():
            rss_deltas.append(proc.memory_info().rss - baseline)
            time.sleep(interval_s)

    thread = threading.Thread(target=sampler, name="tsamd-rss", daemon=True)
    thread.start()
    try:
        yield
    finally:
        stop.set()
        thread.join()


def max_rss_delta_mb(rss_deltas: Deque[int]) -> float:
    return max(rss_deltas, default=0) / _MB
