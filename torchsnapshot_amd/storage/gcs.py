"""Google Cloud Storage backend over aiohttp (JSON/XML API).

No google-cloud libraries ship in this environment, so the plugin talks
to the GCS JSON API directly: resumable uploads in 100 MB chunks for
large objects, simple uploads otherwise, ranged downloads, and a shared
retry strategy with exponential backoff + jitter (parity in capability
with reference torchsnapshot/storage_plugins/gcs.py:141-277).

Auth: a bearer token from storage_options["token"], a callable
storage_options["token_provider"], or the GCS_ACCESS_TOKEN env var
(metadata-server flows need network access this container lacks).
"""

from __future__ import annotations

import asyncio
import random
from typing import Callable, Dict, Optional, Union

from ..io_types import ReadIO, StoragePlugin, WriteIO

_UPLOAD_CHUNK = 100 * 1024 * 1024
_BASE = "https://storage.googleapis.com"


class _RetryStrategy:
    """Shared-deadline retry: the deadline refreshes whenever any transfer
    makes progress, so one slow object doesn't fail a healthy pipeline."""

    def __init__(self, deadline_s: float = 600.0) -> None:
        self.deadline_s = deadline_s
        self._last_progress = None  # type: Optional[float]

    def report_progress(self) -> None:
        self._last_progress = asyncio.get_event_loop().time()

    def out_of_time(self) -> bool:
        if self._last_progress is None:
            self.report_progress()
            return False
        return (
            asyncio.get_event_loop().time() - self._last_progress
            > self.deadline_s
        )

    async def backoff(self, attempt: int) -> None:
        await asyncio.sleep(min(0.5 * 2**attempt, 30.0) * (1 + random.random()))


def _is_transient(status: int) -> bool:
    return status in (408, 429) or status >= 500


class GCSStoragePlugin(StoragePlugin):
    def __init__(self, root: str, storage_options: Optional[dict] = None) -> None:
        opts = storage_options or {}
        bucket, _, prefix = root.partition("/")
        if not bucket:
            raise ValueError(f"invalid gs root: {root!r} (want bucket[/prefix])")
        self.bucket = bucket
        self.prefix = prefix
        self.base = opts.get("endpoint_url", _BASE).rstrip("/")
        token: Union[str, Callable[[], str], None] = opts.get("token")
        if token is None:
            import os

            token = os.environ.get("GCS_ACCESS_TOKEN")
        self._token = token
        self._token_provider: Optional[Callable[[], str]] = opts.get(
            "token_provider"
        )
        self._sessions: Dict[int, object] = {}
        self.retry = _RetryStrategy(float(opts.get("deadline_s", 600.0)))

    def _key(self, path: str) -> str:
        return f"{self.prefix}/{path}" if self.prefix else path

    def _auth_headers(self) -> Dict[str, str]:
        token = self._token_provider() if self._token_provider else self._token
        if not token:
            raise ValueError(
                "GCS auth missing: pass storage_options token/token_provider "
                "or set GCS_ACCESS_TOKEN"
            )
        return {"authorization": f"Bearer {token}"}

    async def _session(self):
        import aiohttp

        loop_id = id(asyncio.get_running_loop())
        sess = self._sessions.get(loop_id)
        if sess is None or sess.closed:
            sess = aiohttp.ClientSession(
                timeout=aiohttp.ClientTimeout(total=1800, connect=60)
            )
            self._sessions[loop_id] = sess
        return sess

    async def write(self, write_io: WriteIO) -> None:
        import urllib.parse

        mv = memoryview(write_io.buf)
        if mv.format != "B":
            mv = mv.cast("B")
        key = urllib.parse.quote(self._key(write_io.path), safe="")
        sess = await self._session()
        if mv.nbytes <= _UPLOAD_CHUNK:
            url = (
                f"{self.base}/upload/storage/v1/b/{self.bucket}/o"
                f"?uploadType=media&name={key}"
            )
            for attempt in range(6):
                async with sess.post(
                    url, data=mv, headers=self._auth_headers()
                ) as resp:
                    if resp.status == 200:
                        self.retry.report_progress()
                        return
                    if not _is_transient(resp.status):
                        raise RuntimeError(
                            f"GCS upload {write_io.path}: {resp.status} "
                            f"{(await resp.text())[:300]}"
                        )
                if self.retry.out_of_time():
                    break
                await self.retry.backoff(attempt)
            raise RuntimeError(f"GCS upload {write_io.path}: retries exhausted")
        # resumable upload for large objects
        start_url = (
            f"{self.base}/upload/storage/v1/b/{self.bucket}/o"
            f"?uploadType=resumable&name={key}"
        )
        async with sess.post(start_url, headers=self._auth_headers()) as resp:
            if resp.status != 200:
                raise RuntimeError(
                    f"GCS resumable init {write_io.path}: {resp.status}"
                )
            session_url = resp.headers["Location"]
        total = mv.nbytes
        off = 0
        while off < total:
            chunk = mv[off : off + _UPLOAD_CHUNK]
            end = off + chunk.nbytes
            headers = {
                "content-length": str(chunk.nbytes),
                "content-range": f"bytes {off}-{end - 1}/{total}",
            }
            for attempt in range(6):
                async with sess.put(session_url, data=chunk, headers=headers) as resp:
                    if resp.status in (200, 201, 308):
                        self.retry.report_progress()
                        break
                    if not _is_transient(resp.status):
                        raise RuntimeError(
                            f"GCS chunk {write_io.path}@{off}: {resp.status}"
                        )
                if self.retry.out_of_time():
                    raise RuntimeError(
                        f"GCS chunk {write_io.path}@{off}: deadline exceeded"
                    )
                await self.retry.backoff(attempt)
            off = end

    async def read(self, read_io: ReadIO) -> None:
        import urllib.parse

        key = urllib.parse.quote(self._key(read_io.path), safe="")
        url = f"{self.base}/storage/v1/b/{self.bucket}/o/{key}?alt=media"
        headers = self._auth_headers()
        if read_io.byte_range is not None:
            start, end = read_io.byte_range
            headers["range"] = f"bytes={start}-{end - 1}"
        sess = await self._session()
        for attempt in range(6):
            async with sess.get(url, headers=headers) as resp:
                if resp.status in (200, 206):
                    read_io.buf = bytearray(await resp.read())
                    self.retry.report_progress()
                    return
                if resp.status == 404:
                    raise FileNotFoundError(f"GCS object missing: {read_io.path}")
                if not _is_transient(resp.status):
                    raise RuntimeError(
                        f"GCS read {read_io.path}: {resp.status}"
                    )
            await self.retry.backoff(attempt)
        raise RuntimeError(f"GCS read {read_io.path}: retries exhausted")

    async def delete(self, path: str) -> None:
        import urllib.parse

        key = urllib.parse.quote(self._key(path), safe="")
        url = f"{self.base}/storage/v1/b/{self.bucket}/o/{key}"
        sess = await self._session()
        async with sess.delete(url, headers=self._auth_headers()) as resp:
            if resp.status not in (200, 204, 404):
                raise RuntimeError(f"GCS delete {path}: {resp.status}")

    # sync one-shot ops (metadata reads etc.) run on throwaway event
    # loops; close the per-loop session inside the same loop so aiohttp
    # connections never leak
    def sync_read(self, read_io) -> None:
        from ..scheduler import run_coro_sync

        async def go():
            try:
                await self.read(read_io)
            finally:
                await self.close()

        run_coro_sync(go())

    def sync_write(self, write_io) -> None:
        from ..scheduler import run_coro_sync

        async def go():
            try:
                await self.write(write_io)
            finally:
                await self.close()

        run_coro_sync(go())

    async def close_for_loop(self) -> None:
        sess = self._sessions.pop(id(asyncio.get_running_loop()), None)
        if sess is not None and not sess.closed:
            await sess.close()

    async def close(self) -> None:
        await self.close_for_loop()
