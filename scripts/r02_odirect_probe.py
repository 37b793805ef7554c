#!/usr/bin/env python3
"""Probe: can O_DIRECT parallel segment reads beat buffered parallel
reads (4.4 GB/s) / the restore path (5.06 GB/s) on this pool's NVMe?
Pure measurement — run on a box with a 16 GB snapshot already on disk.

Usage: python scripts/r02_odirect_probe.py <snapshot_dir>
"""

import ctypes
import mmap
import os
import sys
import time
from concurrent.futures import ThreadPoolExecutor

SEG = 64 * 1024 * 1024
ALIGN = 4096


def aligned_buf(n):
    # mmap gives page-aligned memory
    return mmap.mmap(-1, n)


def jobs_for(root):
    out = []
    for d, _, fs in os.walk(root):
        for f in fs:
            p = os.path.join(d, f)
            size = os.path.getsize(p)
            for off in range(0, size, SEG):
                out.append((p, off, min(off + SEG, size)))
    return out


def read_odirect(job):
    p, s, e = job
    n = e - s
    fd = os.open(p, os.O_RDONLY | os.O_DIRECT)
    try:
        aligned_n = n & ~(ALIGN - 1)
        buf = aligned_buf(max(aligned_n, ALIGN))
        mv = memoryview(buf)
        off = 0
        while off < aligned_n:
            got = os.preadv(fd, [mv[off:aligned_n]], s + off)
            if got == 0:
                break
            off += got
    finally:
        os.close(fd)
    # unaligned tail: buffered
    if n - (n & ~(ALIGN - 1)):
        fd = os.open(p, os.O_RDONLY)
        try:
            tail = bytearray(n - (n & ~(ALIGN - 1)))
            os.preadv(fd, [tail], s + (n & ~(ALIGN - 1)))
        finally:
            os.close(fd)
    return n


def read_buffered(job):
    p, s, e = job
    fd = os.open(p, os.O_RDONLY)
    try:
        os.posix_fadvise(fd, s, e - s, os.POSIX_FADV_SEQUENTIAL)
        os.posix_fadvise(fd, s, e - s, os.POSIX_FADV_WILLNEED)
        buf = bytearray(e - s)
        mv = memoryview(buf)
        off = 0
        while off < len(mv):
            off += os.preadv(fd, [mv[off:]], s + off)
    finally:
        os.close(fd)
    return e - s


def drop_caches():
    os.sync()
    with open("/proc/sys/vm/drop_caches", "w") as f:
        f.write("3")


def run(name, fn, jobs, threads):
    drop_caches()
    t0 = time.monotonic()
    with ThreadPoolExecutor(threads) as ex:
        total = sum(ex.map(fn, jobs))
    dt = time.monotonic() - t0
    print(f"{name} x{threads}: {total/1e9:.2f} GB in {dt:.2f}s = {total/1e9/dt:.2f} GB/s")


def main():
    root = sys.argv[1]
    jobs = jobs_for(root)
    for threads in (8, 16, 32):
        run("odirect", read_odirect, jobs, threads)
    for threads in (16, 32):
        run("buffered", read_buffered, jobs, threads)


if __name__ == "__main__":
    main()
