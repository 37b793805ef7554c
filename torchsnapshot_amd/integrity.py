"""Optional payload integrity checksums (beyond reference parity).

With TSAMD_CHECKSUM=1, every payload buffer is xxh3-hashed as it is
staged and each rank writes a ``<rank>/.checksums`` JSON next to its
payloads (after its writes complete, before the commit barrier — so a
committed snapshot always has complete checksum files). With
TSAMD_VERIFY_CHECKSUM=1, whole-file reads are verified against them on
restore; a mismatch fails the restore loudly instead of loading silently
corrupted weights.

xxh3 runs at >10 GB/s per thread and the hashing happens in the staging
executor, so overhead stays in the noise of storage I/O.
"""

from __future__ import annotations

import json
import logging
import os
from typing import Dict, Optional

from .io_types import ReadIO, StoragePlugin, WriteIO

logger = logging.getLogger(__name__)


def checksumming_enabled() -> bool:
    return os.environ.get("TSAMD_CHECKSUM", "0") not in ("0", "", "false")


def verification_enabled() -> bool:
    return os.environ.get("TSAMD_VERIFY_CHECKSUM", "0") not in ("0", "", "false")


def hash_buffer(buf) -> str:
    import xxhash

    return "xxh3:" + xxhash.xxh3_64_hexdigest(buf)


# -- psum64: the device-side checksum (computed for free inside the gather
# kernel; see ops/hip/staging.hip). An order-independent weighted word sum:
# sum(word_i * (splitmix64(i) | 1)) mod 2^64 over little-endian u64 words,
# zero-padded tail. Verified here vectorized with numpy. ---------------------


def _splitmix64_mult(idx):
    import numpy as np

    z = idx + np.uint64(0x9E3779B97F4A7C15)
    z = (z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
    z = (z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
    z = z ^ (z >> np.uint64(31))
    return z | np.uint64(1)


def psum64_hexdigest(buf, word_base: int = 0) -> str:
    """psum64 of ``buf``. ``word_base`` is the u64 word index of the
    buffer's first byte within the containing file, so subrange checksums
    of byte-range reads line up with the file-global indexing used by the
    device kernel (ops/hip/staging.hip).

    Fast path: a single-pass threaded C++ loop in the native extension
    (memcpy-rate; the vectorized python implementations burn ~10 memory
    passes on splitmix temporaries and manage only ~0.2-0.3 GB/s).
    Falls back to torch int64 ops (two's-complement wraparound IS mod
    2^64 arithmetic) when the extension is unavailable."""
    import torch

    mv = memoryview(buf)
    if mv.format != "B":
        mv = mv.cast("B")
    nbytes = mv.nbytes
    if nbytes == 0:
        return "psum64:" + format(0, "016x")
    try:
        from . import _csnap

        return "psum64:" + format(
            _csnap.psum64_host(mv, word_base) % (1 << 64), "016x"
        )
    except ImportError:
        pass
    total = 0
    chunk_words = 16 * 1024 * 1024  # 128 MB pieces
    off = 0
    wb = word_base
    while off < nbytes:
        end = min(off + chunk_words * 8, nbytes)
        piece = torch.frombuffer(mv[off:end], dtype=torch.uint8)
        pad = (-piece.numel()) % 8
        if pad:
            piece = torch.cat([piece, torch.zeros(pad, dtype=torch.uint8)])
        words = piece.view(torch.int64)
        idx = torch.arange(wb, wb + words.numel(), dtype=torch.int64)
        total = (total + _psum_torch_chunk(words, idx)) % (1 << 64)
        wb += words.numel()
        off = end
    return "psum64:" + format(total, "016x")


def _psum_torch_chunk(words: "torch.Tensor", idx: "torch.Tensor") -> int:
    """sum(words * (splitmix64(idx) | 1)) mod 2^64 with int64 wraparound.
    Right shifts must be LOGICAL (the values are bit patterns, not signed
    numbers): torch >> on int64 is arithmetic, so mask the copied sign
    bits off after shifting."""
    import torch

    def lshr(v: torch.Tensor, n: int) -> torch.Tensor:
        return (v >> n) & ((1 << (64 - n)) - 1)

    def c(val: int) -> int:  # u64 constant -> i64 two's complement
        return val - (1 << 64) if val >= (1 << 63) else val

    z = idx + c(0x9E3779B97F4A7C15)
    z = (z ^ lshr(z, 30)) * c(0xBF58476D1CE4E5B9)
    z = (z ^ lshr(z, 27)) * c(0x94D049BB133111EB)
    z = z ^ lshr(z, 31)
    mult = z | 1
    s = int((words * mult).sum().item())
    return s % (1 << 64)


def checksum_file_path(rank: int) -> str:
    return f"{rank}/.checksums"


def write_checksum_file(
    storage: StoragePlugin, rank: int, checksums: Dict[str, str]
) -> None:
    if not checksums:
        # an unchecksummed save to a path that previously held a
        # checksummed snapshot must not leave the old rank file behind —
        # verification would compare the NEW payloads against the stale
        # values and report phantom corruption
        from .scheduler import run_coro_sync

        async def cleanup() -> None:
            try:
                await storage.delete(checksum_file_path(rank))
            except (FileNotFoundError, OSError):
                pass
            except Exception:
                logger.debug(
                    "could not remove stale checksum file", exc_info=True
                )
            finally:
                try:
                    await storage.close_for_loop()
                except Exception:
                    pass

        run_coro_sync(cleanup())
        return
    storage.sync_write(
        WriteIO(
            path=checksum_file_path(rank),
            buf=json.dumps(checksums, sort_keys=True).encode("utf-8"),
        )
    )


def load_checksums(
    storage: StoragePlugin, world_size: int
) -> Optional[Dict[str, str]]:
    """Merged {payload_path: xxh3} for all writer ranks, or None if the
    snapshot was taken without checksumming."""
    merged: Dict[str, str] = {}
    found = False
    for rank in range(world_size):
        read_io = ReadIO(path=checksum_file_path(rank))
        try:
            storage.sync_read(read_io)
        except FileNotFoundError:
            continue
        found = True
        merged.update(json.loads(bytes(read_io.buf).decode("utf-8")))
    return merged if found else None


def member_key(path: str, start: int, end: int) -> str:
    """Checksum-file key for one batched-slab member's byte range."""
    return f"{path}#{start}-{end}"


def len_key(path: str) -> str:
    """Checksum-file key recording a payload file's byte length (lets
    restore verify a file read as byte-range tiles once the tiles cover
    the whole file)."""
    return f"{path}#len"


def psum64_value(buf, word_base: int = 0) -> int:
    """psum64 as an integer (for accumulating partial sums)."""
    return int(psum64_hexdigest(buf, word_base)[len("psum64:"):], 16)


def expected_span_psum(
    path: str, byte_range, expected: Dict[str, str]
) -> Optional[int]:
    """Expected psum64 of the file span ``byte_range`` of ``path``: the
    (mod 2^64) sum of every recorded member checksum inside it (slab
    padding between members is zeroed and contributes nothing). None when
    no member of the span was recorded."""
    start, end = byte_range
    prefix = path + "#"
    total = 0
    found = False
    for k, v in expected.items():
        if not k.startswith(prefix) or not v.startswith("psum64:"):
            continue
        s_str, _, e_str = k[len(prefix):].partition("-")
        if not (s_str.isdigit() and e_str.isdigit()):
            continue
        s, e = int(s_str), int(e_str)
        if s >= start and e <= end:
            total = (total + int(v[len("psum64:"):], 16)) % (1 << 64)
            found = True
    return total if found else None


def verify_ranged_buffer(
    path: str, buf, byte_range, expected: Dict[str, str]
) -> bool:
    """Verify a byte-range read (a batched-slab member or a merged span of
    members) against per-member psum64 values recorded at save time.
    Returns True when verification happened, False when the range had no
    recorded members (the caller may then fall back to whole-file
    partial-sum accumulation); raises on mismatch.

    psum64 is additive over disjoint file word-ranges and slab padding is
    zeroed, so the expected checksum of any member-aligned span is the
    (mod 2^64) sum of the recorded per-member values inside it.
    """
    start, end = byte_range
    total = expected_span_psum(path, byte_range, expected)
    if total is None:
        return False
    if start % 8 != 0:
        # slab members are 64-byte aligned; a misaligned span can't use
        # file-global word indexing — don't verify rather than misreport
        logger.warning(
            "skipping checksum verification of misaligned span %s[%d:%d]",
            path, start, end,
        )
        return False
    got = psum64_hexdigest(buf, word_base=start // 8)
    want = "psum64:" + format(total, "016x")
    if got != want:
        raise RuntimeError(
            f"checksum mismatch for span '{path}' [{start}:{end}): snapshot "
            f"recorded {want}, read back {got} — the file is corrupted or "
            "was modified after the snapshot was committed"
        )
    return True


def verify_buffer(path: str, buf, expected: Dict[str, str]) -> None:
    want = expected.get(path)
    if want is None:
        return
    if want.startswith("psum64:"):
        got = psum64_hexdigest(buf)
    else:
        got = hash_buffer(buf)
        if not want.startswith("xxh3:"):
            got = got.removeprefix("xxh3:")  # legacy untagged
    if got != want:
        raise RuntimeError(
            f"checksum mismatch for payload '{path}': snapshot recorded "
            f"{want}, read back {got} — the file is corrupted or was "
            "modified after the snapshot was committed"
        )
