"""async_take: early resume, atomic commit, fault injection.

Fault injection mirrors the reference pattern (tests/test_async_take.py:
27-66): storage subclasses that sleep or raise, injected by patching the
plugin resolver; a failed snapshot must not commit .snapshot_metadata."""

import asyncio
import os
import tempfile
import time
from unittest import mock

import pytest
import torch

from torchsnapshot_amd import Snapshot, StateDict
from torchsnapshot_amd.storage.fs import FSStoragePlugin
from torchsnapshot_amd.test_utils import check_state_dict_eq


class SlowFSStoragePlugin(FSStoragePlugin):
    async def write(self, write_io) -> None:
        await asyncio.sleep(0.3)
        await super().write(write_io)


class FaultyFSStoragePlugin(FSStoragePlugin):
    async def write(self, write_io) -> None:
        if write_io.path != ".snapshot_metadata":
            raise RuntimeError("injected storage failure")
        await super().write(write_io)


def _patch_plugin(cls):
    def fake(url, storage_options=None):
        path = url.split("://")[-1]
        return cls(path, storage_options)

    return mock.patch(
        "torchsnapshot_amd.snapshot.url_to_storage_plugin", side_effect=fake
    )


def test_async_take_returns_before_io_done():
    sd = StateDict(big=torch.rand(512, 512), small=torch.rand(10))
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "snap")
        with _patch_plugin(SlowFSStoragePlugin):
            t0 = time.monotonic()
            pending = Snapshot.async_take(path, {"sd": sd})
            returned_after = time.monotonic() - t0
            assert not pending.done() or returned_after < 10
            snapshot = pending.wait()
        sd2 = StateDict()
        snapshot.restore({"sd": sd2})
        assert check_state_dict_eq(sd.state_dict(), sd2.state_dict())


def test_async_take_mutation_after_return_is_safe():
    sd = StateDict(w=torch.rand(256, 256))
    saved = sd["w"].clone()
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "snap")
        with _patch_plugin(SlowFSStoragePlugin):
            pending = Snapshot.async_take(path, {"sd": sd})
            # training mutates the tensor while I/O is still in flight
            sd["w"].fill_(-1.0)
            snapshot = pending.wait()
        out = StateDict()
        snapshot.restore({"sd": out})
        assert torch.equal(out["w"], saved)


def test_failed_async_take_commits_no_metadata():
    sd = StateDict(w=torch.rand(64))
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "snap")
        with _patch_plugin(FaultyFSStoragePlugin):
            # the injected error may surface at async_take (if I/O fails
            # before staging completes) or at wait(); both are valid — what
            # matters is that no metadata is committed
            with pytest.raises(RuntimeError):
                pending = Snapshot.async_take(path, {"sd": sd})
                pending.wait()
        assert not os.path.exists(os.path.join(path, ".snapshot_metadata"))


def test_failed_sync_take_commits_no_metadata():
    sd = StateDict(w=torch.rand(64))
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "snap")
        with _patch_plugin(FaultyFSStoragePlugin):
            with pytest.raises(RuntimeError):
                Snapshot.take(path, {"sd": sd})
        assert not os.path.exists(os.path.join(path, ".snapshot_metadata"))


def _async_take_dist(tmpdir: str) -> None:
    import os

    import torch.distributed as dist

    from torchsnapshot_amd.test_utils import check_state_dict_eq

    torch.manual_seed(10 + dist.get_rank())
    sd = StateDict(
        mine=torch.rand(64, 32),
        shared=torch.full((16,), float(dist.get_rank())),
    )
    path = os.path.join(tmpdir, "snap")
    pending = Snapshot.async_take(path, {"sd": sd})
    snapshot = pending.wait()
    assert os.path.exists(os.path.join(path, ".snapshot_metadata"))
    out = StateDict(mine=torch.zeros(64, 32), shared=torch.zeros(16))
    snapshot.restore({"sd": out})
    assert torch.equal(out["mine"], sd["mine"])
    assert torch.equal(out["shared"], sd["shared"])


def test_async_take_world2():
    import tempfile as tf

    from torchsnapshot_amd.test_utils import run_multiprocess

    with tf.TemporaryDirectory() as d:
        run_multiprocess(2, _async_take_dist, d)


def test_two_concurrent_async_takes():
    """Two async snapshots of different app states to different paths may
    overlap; both must commit correctly."""
    sd1 = StateDict(a=torch.rand(128, 64))
    sd2 = StateDict(b=torch.rand(256, 16))
    with tempfile.TemporaryDirectory() as d:
        with _patch_plugin(SlowFSStoragePlugin):
            p1 = Snapshot.async_take(os.path.join(d, "s1"), {"sd": sd1})
            p2 = Snapshot.async_take(os.path.join(d, "s2"), {"sd": sd2})
            s1 = p1.wait()
            s2 = p2.wait()
        out1, out2 = StateDict(), StateDict()
        s1.restore({"sd": out1})
        s2.restore({"sd": out2})
        assert torch.equal(out1["a"], sd1["a"])
        assert torch.equal(out2["b"], sd2["b"])


def _async_take_peer_failure(tmpdir: str) -> None:
    """Rank 1's storage fails mid-async-drain: rank 0 must observe the
    error through the store barrier, raise from wait(), and commit no
    metadata."""
    import os

    import torch.distributed as dist

    sd = StateDict(w=torch.rand(64, 64))
    path = os.path.join(tmpdir, "snap")
    if dist.get_rank() == 1:
        with _patch_plugin(FaultyFSStoragePlugin):
            with pytest.raises(RuntimeError):
                pending = Snapshot.async_take(path, {"sd": sd})
                pending.wait()
    else:
        pending = Snapshot.async_take(path, {"sd": sd})
        with pytest.raises(RuntimeError):
            pending.wait()
    dist.barrier()
    assert not os.path.exists(os.path.join(path, ".snapshot_metadata"))


def test_async_take_peer_failure_world2():
    import tempfile as tf

    from torchsnapshot_amd.test_utils import run_multiprocess

    with tf.TemporaryDirectory() as d:
        run_multiprocess(2, _async_take_peer_failure, d)


def test_async_take_custom_tensor_prepare_func():
    """async_take accepts the save-time tensor transform too (parity with
    reference snapshot.py:230-293)."""
    sd = StateDict(w=torch.rand(64, 32))
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "snap")
        pending = Snapshot.async_take(
            path,
            {"sd": sd},
            _custom_tensor_prepare_func=lambda p, t: t.to(torch.bfloat16),
        )
        snap = pending.wait()
        assert snap.get_manifest()["0/sd/w"]["dtype"] == "bfloat16"
        out = StateDict(w=torch.zeros(64, 32))
        snap.restore({"sd": out})
        assert torch.equal(out["w"], sd["w"].to(torch.bfloat16).float())


def test_async_shadow_returns_without_staging_wait(monkeypatch):
    """With shadow clones, async_take must not block on staging: inject a
    slow stager-side storage and check the return races ahead of I/O
    while mutation safety still holds."""
    monkeypatch.setenv("TSAMD_ASYNC_SHADOW", "1")
    sd = StateDict(w=torch.rand(256, 256), b=torch.rand(64))
    saved_w = sd["w"].clone()
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "snap")
        with _patch_plugin(SlowFSStoragePlugin):
            pending = Snapshot.async_take(path, {"sd": sd})
            assert pending.sources_immutable
            sd["w"].fill_(-123.0)  # mutate immediately
            snap = pending.wait()
        out = StateDict()
        snap.restore({"sd": out})
        assert torch.equal(out["w"], saved_w)


def test_async_shadow_falls_back_on_unshadowable_leaves(monkeypatch):
    """Objects that can't be cloned safely force the classic
    wait-for-staging path (sources_immutable False)."""
    monkeypatch.setenv("TSAMD_ASYNC_SHADOW", "1")
    # a set is a picklable non-tensor LEAF (flatten doesn't descend into
    # it), so it lands in the object preparer and can't be shadowed
    sd = StateDict(w=torch.rand(16), blob={1, 2, 3})
    with tempfile.TemporaryDirectory() as d:
        pending = Snapshot.async_take(os.path.join(d, "snap"), {"sd": sd})
        assert not pending.sources_immutable
        pending.wait()


def test_async_shadow_tied_weights_still_dedup(monkeypatch):
    monkeypatch.setenv("TSAMD_ASYNC_SHADOW", "1")
    shared = torch.rand(128)

    class Tied:
        def state_dict(self):
            return {"a": shared, "b": shared}

        def load_state_dict(self, sd):
            pass

    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "snap")
        pending = Snapshot.async_take(path, {"m": Tied()})
        snap = pending.wait()
        man = snap.get_manifest()
        # both paths point at ONE payload (same location + byte range)
        a, b = man["0/m/a"], man["0/m/b"]
        assert (a.get("location"), a.get("byte_range")) == (
            b.get("location"),
            b.get("byte_range"),
        )
        out = {}

        class Out:
            def state_dict(self):
                return {"a": torch.zeros(128), "b": torch.zeros(128)}

            def load_state_dict(self, sd):
                out.update(sd)

        snap.restore({"m": Out()})
        assert torch.equal(out["a"], shared)
        assert torch.equal(out["b"], shared)
