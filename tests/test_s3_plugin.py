"""S3 plugin tests against an in-process fake S3 (aiohttp server on
127.0.0.1): auth header shape, byte ranges, and a full snapshot
take/restore through the plugin."""

import asyncio
import threading

import pytest
import torch

from torchsnapshot_amd import Snapshot, StateDict
from torchsnapshot_amd.test_utils import check_state_dict_eq

aiohttp = pytest.importorskip("aiohttp")
from aiohttp import web  # noqa: E402

pytestmark = pytest.mark.timeout(120)


class FakeS3:
    def __init__(self):
        self.objects = {}
        self.auth_headers = []
        self.port = None
        self._started = threading.Event()
        self._stop = None
        self.thread = threading.Thread(target=self._run, daemon=True)
        self.thread.start()
        assert self._started.wait(20)

    def _run(self):
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
        self._stop = loop.create_future()

        async def handler(request: web.Request):
            key = request.match_info["key"]
            self.auth_headers.append(request.headers.get("Authorization", ""))
            if request.method == "PUT":
                self.objects[key] = await request.read()
                return web.Response(status=200)
            if request.method == "GET" and "list-type" in request.query:
                prefix = request.query.get("prefix", "")
                # path-style: request path is the bucket; stored keys are
                # "<bucket>/<key>" but real S3 lists keys without bucket
                bucket = key.split("/")[0] if key else ""
                in_bucket = [
                    k[len(bucket) + 1 :]
                    for k in self.objects
                    if k.startswith(bucket + "/")
                ]
                keys = [k for k in in_bucket if k.startswith(prefix)]
                body = "<ListBucketResult>" + "".join(
                    f"<Key>{k}</Key>" for k in sorted(keys)
                ) + "</ListBucketResult>"
                return web.Response(status=200, text=body)
            if request.method == "GET":
                if key not in self.objects:
                    return web.Response(status=404)
                data = self.objects[key]
                rng = request.headers.get("Range")
                if rng:
                    spec = rng.split("=")[1]
                    start, end = spec.split("-")
                    data = data[int(start) : int(end) + 1]
                    return web.Response(status=206, body=data)
                return web.Response(status=200, body=data)
            if request.method == "DELETE":
                self.objects.pop(key, None)
                return web.Response(status=204)
            return web.Response(status=400)

        async def main():
            app = web.Application(client_max_size=1024**3)
            app.router.add_route(
                "*", "/{key:.*}", handler
            )
            runner = web.AppRunner(app)
            await runner.setup()
            site = web.TCPSite(runner, "127.0.0.1", 0)
            await site.start()
            self.port = runner.addresses[0][1]
            self._started.set()
            await self._stop
            await runner.cleanup()

        loop.run_until_complete(main())

    def stop(self):
        pass  # daemon thread; test process exit cleans up


@pytest.fixture(scope="module")
def fake_s3():
    server = FakeS3()
    yield server
    server.stop()


def _options(server):
    return {
        "endpoint_url": f"http://127.0.0.1:{server.port}",
        "access_key_id": "test-ak",
        "secret_access_key": "test-sk",
        "region": "us-test-1",
    }


def test_s3_snapshot_round_trip(fake_s3):
    sd = StateDict(
        w=torch.rand(128, 64),
        small=torch.rand(5),
        n=17,
    )
    snapshot = Snapshot.take(
        "s3://bkt/ckpt", {"sd": sd}, storage_options=_options(fake_s3)
    )
    # payloads + metadata landed under the bucket/prefix
    assert any(k.startswith("bkt/ckpt/") for k in fake_s3.objects)
    assert "bkt/ckpt/.snapshot_metadata" in fake_s3.objects
    # SigV4 headers present
    assert any(h.startswith("AWS4-HMAC-SHA256") for h in fake_s3.auth_headers)

    sd2 = StateDict()
    snapshot2 = Snapshot(
        "s3://bkt/ckpt", storage_options=_options(fake_s3)
    )
    snapshot2.restore({"sd": sd2})
    assert check_state_dict_eq(sd.state_dict(), sd2.state_dict())


def test_s3_byte_range_read(fake_s3):
    sd = StateDict(big=torch.rand(1000, 100))
    Snapshot.take(
        "s3://bkt/rng", {"sd": sd}, storage_options=_options(fake_s3)
    )
    snap = Snapshot("s3://bkt/rng", storage_options=_options(fake_s3))
    out = snap.read_object("0/sd/big", memory_budget_bytes=64 * 1024)
    assert torch.equal(out, sd["big"])


def test_s3_missing_creds():
    import os

    env_backup = {
        k: os.environ.pop(k, None)
        for k in ("AWS_ACCESS_KEY_ID", "AWS_SECRET_ACCESS_KEY")
    }
    try:
        from torchsnapshot_amd.storage.s3 import S3StoragePlugin

        with pytest.raises(ValueError, match="credentials"):
            S3StoragePlugin("bucket/prefix", {})
    finally:
        for k, v in env_backup.items():
            if v is not None:
                os.environ[k] = v


def test_s3_delete_snapshot(fake_s3):
    sd = StateDict(w=torch.rand(32))
    snap = Snapshot.take(
        "s3://bkt/del", {"sd": sd}, storage_options=_options(fake_s3)
    )
    assert any(k.startswith("bkt/del/") for k in fake_s3.objects)
    snap.delete()
    assert not any(k.startswith("bkt/del/") for k in fake_s3.objects)
