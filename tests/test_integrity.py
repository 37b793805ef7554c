"""Payload checksum tests: written on take, verified on restore,
corruption detected."""

import os

import pytest
import torch

from torchsnapshot_amd import Snapshot, StateDict
from torchsnapshot_amd.test_utils import tmp_snapshot_path


def test_checksums_written_and_verified(monkeypatch):
    monkeypatch.setenv("TSAMD_CHECKSUM", "1")
    monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "1")
    sd = StateDict(a=torch.rand(128, 16), b=torch.rand(64))
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        assert os.path.exists(os.path.join(path, "0", ".checksums"))
        out = StateDict()
        snap.restore({"sd": out})
        assert torch.equal(out["a"], sd["a"])


def test_corruption_detected(monkeypatch):
    monkeypatch.setenv("TSAMD_CHECKSUM", "1")
    monkeypatch.setenv("TSAMD_DISABLE_BATCHING", "1")
    sd = StateDict(a=torch.rand(256, 64))
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        payload = os.path.join(path, "0", "sd", "a")
        assert os.path.exists(payload)
        # flip one byte
        with open(payload, "r+b") as f:
            f.seek(100)
            orig = f.read(1)
            f.seek(100)
            f.write(bytes([orig[0] ^ 0xFF]))
        monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "1")
        with pytest.raises(RuntimeError, match="checksum mismatch"):
            snap.restore({"sd": StateDict(a=torch.zeros(256, 64))})
        # without verification the (corrupt) load goes through silently,
        # which is exactly why the knob exists
        monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "0")
        snap.restore({"sd": StateDict(a=torch.zeros(256, 64))})


def test_no_checksums_ok(monkeypatch):
    # verifying a snapshot taken without checksums is a no-op
    monkeypatch.delenv("TSAMD_CHECKSUM", raising=False)
    sd = StateDict(a=torch.rand(16))
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        assert not os.path.exists(os.path.join(path, "0", ".checksums"))
        monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "1")
        out = StateDict()
        snap.restore({"sd": out})
        assert torch.equal(out["a"], sd["a"])


def test_async_take_checksums(monkeypatch):
    monkeypatch.setenv("TSAMD_CHECKSUM", "1")
    sd = StateDict(a=torch.rand(64, 64))
    with tmp_snapshot_path() as path:
        pending = Snapshot.async_take(path, {"sd": sd})
        pending.wait()
        assert os.path.exists(os.path.join(path, "0", ".checksums"))


def _checksummed_dist(tmpdir: str) -> None:
    import os

    import torch.distributed as dist

    os.environ["TSAMD_CHECKSUM"] = "1"
    os.environ["TSAMD_VERIFY_CHECKSUM"] = "1"
    torch.manual_seed(5)
    shared = torch.rand(64, 8)
    sd = StateDict(
        shared=shared.clone(),
        mine=torch.full((8,), float(dist.get_rank())),
    )
    path = os.path.join(tmpdir, "snap")
    Snapshot.take(path, {"sd": sd}, replicated=["sd/shared"])
    # each writer rank produced its checksum file
    assert any(
        os.path.exists(os.path.join(path, str(r), ".checksums"))
        for r in range(dist.get_world_size())
    )
    out = StateDict(shared=torch.zeros(64, 8), mine=torch.zeros(8))
    Snapshot(path).restore({"sd": out})
    assert torch.equal(out["shared"], shared)
    assert torch.equal(out["mine"], torch.full((8,), float(dist.get_rank())))


def test_checksums_distributed_world2():
    import tempfile

    from torchsnapshot_amd.test_utils import run_multiprocess

    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _checksummed_dist, d)


def test_batched_member_corruption_detected(monkeypatch):
    """Byte-range reads of batched slabs are verified against per-member
    psum64 checksums (round-1 advisor: they were silently unverified)."""
    monkeypatch.setenv("TSAMD_CHECKSUM", "1")
    # several small tensors -> one batched slab with byte-ranged members
    sd = StateDict(**{f"t{i}": torch.rand(128) for i in range(6)})
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        slab = os.path.join(path, "batched", "0-0")
        assert os.path.exists(slab), "expected a batched slab"
        import json

        cks = json.load(open(os.path.join(path, "0", ".checksums")))
        assert any("#" in k for k in cks), "expected per-member checksums"
        with open(slab, "r+b") as f:
            f.seek(3)
            orig = f.read(1)
            f.seek(3)
            f.write(bytes([orig[0] ^ 0x55]))
        monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "1")
        with pytest.raises(RuntimeError, match="checksum mismatch"):
            snap.restore(
                {"sd": StateDict(**{f"t{i}": torch.zeros(128) for i in range(6)})}
            )


def test_batched_members_verify_clean(monkeypatch):
    monkeypatch.setenv("TSAMD_CHECKSUM", "1")
    monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "1")
    sd = StateDict(**{f"t{i}": torch.rand(64, 3) for i in range(5)})
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict(**{f"t{i}": torch.zeros(64, 3) for i in range(5)})
        snap.restore({"sd": out})
        for i in range(5):
            assert torch.equal(out[f"t{i}"], sd[f"t{i}"])


def test_tiled_read_verification(monkeypatch):
    """A file read as byte-range tiles (read_object with a memory budget)
    is verified once the tiles cover the whole file — and corruption in
    ANY tile fails the read (psum64 whole-file accumulation)."""
    monkeypatch.setenv("TSAMD_CHECKSUM", "1")
    monkeypatch.setenv("TSAMD_DISABLE_BATCHING", "1")
    sd = StateDict(big=torch.rand(512, 256))  # 512 KB single payload
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        import json

        cks = json.load(open(os.path.join(path, "0", ".checksums")))
        # every payload records psum64 (host-staged included), so tiled
        # accumulation works out of the box
        assert cks["0/sd/big"].startswith("psum64:")
        assert cks["0/sd/big#len"]
        payload = os.path.join(path, "0", "sd", "big")

        monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "1")
        out = snap.read_object("0/sd/big", memory_budget_bytes=64 * 1024)
        assert torch.equal(out, sd["big"])

        # corrupt one byte deep inside the file: a tiled read must fail
        with open(payload, "r+b") as f:
            f.seek(200 * 1024)
            b = f.read(1)
            f.seek(200 * 1024)
            f.write(bytes([b[0] ^ 0x1]))
        with pytest.raises(RuntimeError, match="tiled read"):
            snap.read_object("0/sd/big", memory_budget_bytes=64 * 1024)


def test_stale_checksum_file_cleared_on_unchecksummed_resave(monkeypatch):
    """Re-saving WITHOUT checksums to a path that previously held a
    checksummed snapshot must clear the old .checksums file — otherwise
    verification compares new payloads against stale values and reports
    phantom corruption."""
    sd = StateDict(a=torch.rand(256, 16))
    with tmp_snapshot_path() as path:
        monkeypatch.setenv("TSAMD_CHECKSUM", "1")
        Snapshot.take(path, {"sd": sd})
        assert os.path.exists(os.path.join(path, "0", ".checksums"))

        monkeypatch.setenv("TSAMD_CHECKSUM", "0")
        sd2 = StateDict(a=torch.rand(256, 16))  # different content
        snap = Snapshot.take(path, {"sd": sd2})
        assert not os.path.exists(os.path.join(path, "0", ".checksums"))

        monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "1")
        out = StateDict()
        snap.restore({"sd": out})  # no phantom mismatch
        assert torch.equal(out["a"], sd2["a"])
