"""GCS plugin tests against an in-process fake of the GCS JSON API
(simple + resumable uploads, ranged download)."""

import asyncio
import threading

import pytest
import torch

from torchsnapshot_amd import Snapshot, StateDict
from torchsnapshot_amd.test_utils import check_state_dict_eq

aiohttp = pytest.importorskip("aiohttp")
from aiohttp import web  # noqa: E402

pytestmark = pytest.mark.timeout(120)


class FakeGCS:
    def __init__(self):
        self.objects = {}
        self.resumable = {}
        self.port = None
        self.bearer_seen = []
        self._started = threading.Event()
        self.thread = threading.Thread(target=self._run, daemon=True)
        self.thread.start()
        assert self._started.wait(20)

    def _run(self):
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)

        async def upload(request: web.Request):
            self.bearer_seen.append(request.headers.get("Authorization", ""))
            bucket = request.match_info["bucket"]
            utype = request.query.get("uploadType")
            name = request.query.get("name")
            if utype == "media":
                self.objects[f"{bucket}/{name}"] = await request.read()
                return web.json_response({"name": name})
            if utype == "resumable":
                sid = f"sess-{len(self.resumable)}"
                self.resumable[sid] = (f"{bucket}/{name}", bytearray())
                return web.json_response(
                    {},
                    headers={
                        "Location": f"http://127.0.0.1:{self.port}/resume/{sid}"
                    },
                )
            return web.Response(status=400)

        async def resume_put(request: web.Request):
            sid = request.match_info["sid"]
            key, buf = self.resumable[sid]
            data = await request.read()
            crange = request.headers.get("Content-Range", "")
            buf.extend(data)
            total = crange.rsplit("/", 1)[-1]
            if total != "*" and len(buf) >= int(total):
                self.objects[key] = bytes(buf)
                return web.Response(status=200)
            return web.Response(status=308)

        async def download(request: web.Request):
            bucket = request.match_info["bucket"]
            name = request.match_info["key"]
            import urllib.parse

            key = f"{bucket}/{urllib.parse.unquote(name)}"
            if request.query.get("alt") != "media":
                return web.Response(status=400)
            if key not in self.objects:
                return web.Response(status=404)
            data = self.objects[key]
            rng = request.headers.get("Range")
            if rng:
                spec = rng.split("=")[1]
                start, end = spec.split("-")
                return web.Response(
                    status=206, body=data[int(start) : int(end) + 1]
                )
            return web.Response(status=200, body=data)

        async def delete(request: web.Request):
            import urllib.parse

            bucket = request.match_info["bucket"]
            key = f"{bucket}/{urllib.parse.unquote(request.match_info['key'])}"
            self.objects.pop(key, None)
            return web.Response(status=204)

        async def main():
            app = web.Application(client_max_size=1024**3)
            app.router.add_post("/upload/storage/v1/b/{bucket}/o", upload)
            app.router.add_put("/resume/{sid}", resume_put)
            app.router.add_get("/storage/v1/b/{bucket}/o/{key}", download)
            app.router.add_delete("/storage/v1/b/{bucket}/o/{key}", delete)
            runner = web.AppRunner(app)
            await runner.setup()
            site = web.TCPSite(runner, "127.0.0.1", 0)
            await site.start()
            self.port = runner.addresses[0][1]
            self._started.set()
            await loop.create_future()

        loop.run_until_complete(main())


@pytest.fixture(scope="module")
def fake_gcs():
    yield FakeGCS()


def _options(server):
    return {
        "endpoint_url": f"http://127.0.0.1:{server.port}",
        "token": "fake-token",
    }


def test_gcs_snapshot_round_trip(fake_gcs):
    sd = StateDict(w=torch.rand(64, 32), n=5)
    snapshot = Snapshot.take(
        "gs://bkt/ckpt", {"sd": sd}, storage_options=_options(fake_gcs)
    )
    assert "bkt/ckpt/.snapshot_metadata" in fake_gcs.objects
    assert any(b.startswith("Bearer") for b in fake_gcs.bearer_seen)
    out = StateDict()
    Snapshot("gs://bkt/ckpt", storage_options=_options(fake_gcs)).restore(
        {"sd": out}
    )
    assert check_state_dict_eq(sd.state_dict(), out.state_dict())


def test_gcs_byte_range(fake_gcs):
    sd = StateDict(big=torch.rand(500, 100))
    Snapshot.take("gs://bkt/rng", {"sd": sd}, storage_options=_options(fake_gcs))
    snap = Snapshot("gs://bkt/rng", storage_options=_options(fake_gcs))
    out = snap.read_object("0/sd/big", memory_budget_bytes=32 * 1024)
    assert torch.equal(out, sd["big"])


def test_gcs_resumable_upload(fake_gcs, monkeypatch):
    import torchsnapshot_amd.storage.gcs as gcs_mod

    monkeypatch.setattr(gcs_mod, "_UPLOAD_CHUNK", 64 * 1024)
    sd = StateDict(big=torch.rand(300, 100))  # 120 KB -> 2 chunks
    Snapshot.take(
        "gs://bkt/res", {"sd": sd}, storage_options=_options(fake_gcs)
    )
    out = StateDict()
    Snapshot("gs://bkt/res", storage_options=_options(fake_gcs)).restore(
        {"sd": out}
    )
    assert torch.equal(out["big"], sd["big"])


def test_gcs_missing_token():
    from torchsnapshot_amd.storage.gcs import GCSStoragePlugin

    import os

    env = os.environ.pop("GCS_ACCESS_TOKEN", None)
    try:
        plugin = GCSStoragePlugin("bucket/p", {})
        with pytest.raises(ValueError, match="auth missing"):
            plugin._auth_headers()
    finally:
        if env is not None:
            os.environ["GCS_ACCESS_TOKEN"] = env
