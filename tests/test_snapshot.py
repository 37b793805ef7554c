"""End-to-end single-process take/restore tests (CPU)."""

import os

import pytest
import torch

from torchsnapshot_amd import RNGState, Snapshot, StateDict
from torchsnapshot_amd.test_utils import (
    check_state_dict_eq,
    rand_tensor,
    tmp_snapshot_path,
)


class _Model(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.lin1 = torch.nn.Linear(17, 33)
        self.lin2 = torch.nn.Linear(33, 5)
        self.register_buffer("buf", torch.rand(8))

    def forward(self, x):
        return self.lin2(torch.relu(self.lin1(x)))


def test_take_restore_module(toggle_batching):
    model = _Model()
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"model": model})
        assert os.path.exists(os.path.join(path, ".snapshot_metadata"))
        model2 = _Model()
        assert not check_state_dict_eq(model.state_dict(), model2.state_dict())
        snapshot.restore({"model": model2})
        assert check_state_dict_eq(model.state_dict(), model2.state_dict())


def test_take_restore_module_and_optimizer():
    model = _Model()
    optim = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    # generate optimizer state
    model(torch.rand(4, 17)).sum().backward()
    optim.step()
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"model": model, "optim": optim})
        model2 = _Model()
        optim2 = torch.optim.SGD(model2.parameters(), lr=0.1, momentum=0.9)
        model2(torch.rand(4, 17)).sum().backward()
        optim2.step()
        snapshot.restore({"model": model2, "optim": optim2})
        assert check_state_dict_eq(model.state_dict(), model2.state_dict())
        assert check_state_dict_eq(optim.state_dict(), optim2.state_dict())


def test_primitives_and_objects():
    progress = StateDict(
        epoch=7,
        step=123,
        lr=0.125,
        name="run-42",
        done=False,
        blob=b"\x00\x01",
        nested={"a": [1, 2, 3], "t": torch.arange(5)},
        custom=(1, 2),  # tuple -> object fallback
    )
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"progress": progress})
        progress2 = StateDict()
        snapshot.restore({"progress": progress2})
        assert progress2["epoch"] == 7
        assert progress2["step"] == 123
        assert progress2["lr"] == 0.125
        assert progress2["name"] == "run-42"
        assert progress2["done"] is False
        assert progress2["blob"] == b"\x00\x01"
        assert torch.equal(progress2["nested"]["t"], torch.arange(5))
        assert progress2["nested"]["a"] == [1, 2, 3]
        assert progress2["custom"] == (1, 2)


def test_all_dtypes(toggle_batching):
    dtypes = [
        torch.float32,
        torch.float64,
        torch.float16,
        torch.bfloat16,
        torch.complex64,
        torch.uint8,
        torch.int8,
        torch.int16,
        torch.int32,
        torch.int64,
        torch.bool,
        torch.qint8,
        torch.quint8,
    ]
    sd = StateDict({str(d): rand_tensor((7, 3), d) for d in dtypes})
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"sd": sd})
        sd2 = StateDict()
        snapshot.restore({"sd": sd2})
        assert check_state_dict_eq(sd.state_dict(), sd2.state_dict())


def test_chunked_tensor(toggle_chunking):
    big = torch.rand(1000, 10)
    sd = StateDict(big=big)
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"sd": sd})
        sd2 = StateDict(big=torch.zeros(1000, 10))
        snapshot.restore({"sd": sd2})
        assert torch.equal(sd2["big"], big)


def test_rng_state_invariance():
    torch.manual_seed(777)
    rng = RNGState()
    with tmp_snapshot_path() as path:
        state_before = torch.get_rng_state()
        snapshot = Snapshot.take(path, {"rng": rng, "x": StateDict(v=1)})
        # taking the snapshot must not perturb the RNG stream
        assert torch.equal(torch.get_rng_state(), state_before)
        draw_after_take = torch.rand(3)
        # restore, then the same draw must reproduce
        snapshot.restore({"rng": rng})
        assert torch.equal(torch.rand(3), draw_after_take)


def test_restore_into_different_dtype():
    sd = StateDict(t=torch.rand(5, dtype=torch.float32))
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"sd": sd})
        sd2 = StateDict(t=torch.zeros(5, dtype=torch.float64))
        snapshot.restore({"sd": sd2})
        assert torch.allclose(sd2["t"].float(), sd["t"])
        assert sd2["t"].dtype == torch.float64


def test_metadata_missing_raises():
    with tmp_snapshot_path() as path:
        os.makedirs(path, exist_ok=True)
        with pytest.raises(RuntimeError, match="incomplete or corrupted"):
            _ = Snapshot(path).metadata


def test_get_manifest():
    sd = StateDict(a=1, t=torch.rand(3))
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"sd": sd})
        manifest = snapshot.get_manifest()
        assert "0/sd/a" in manifest
        assert manifest["0/sd/t"]["kind"] == "tensor"


def test_non_contiguous_and_view_tensors():
    base = torch.rand(10, 10)
    sd = StateDict(
        transposed=base.t(),
        narrow=base[2:5],
        strided=base[::2, ::3],
        expanded=torch.rand(1, 4).expand(3, 4),
    )
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"sd": sd})
        sd2 = StateDict(
            transposed=torch.zeros(10, 10),
            narrow=torch.zeros(3, 10),
            strided=torch.zeros(5, 4),
            expanded=torch.zeros(3, 4),
        )
        snapshot.restore({"sd": sd2})
        assert torch.equal(sd2["transposed"], base.t())
        assert torch.equal(sd2["narrow"], base[2:5])
        assert torch.equal(sd2["strided"], base[::2, ::3])
        assert torch.equal(sd2["expanded"], torch.rand(1, 4).expand(3, 4) * 0 + sd["expanded"])


def test_read_object():
    sd = StateDict(t=torch.rand(17), n=42)
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"sd": sd})
        t = snapshot.read_object("0/sd/t")
        assert torch.equal(t, sd["t"])
        assert snapshot.read_object("0/sd/n") == 42
        out = torch.zeros(17)
        snapshot.read_object("0/sd/t", obj_out=out)
        assert torch.equal(out, sd["t"])


def test_read_object_tiled():
    sd = StateDict(t=torch.rand(1000, 100))
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"sd": sd})
        t = snapshot.read_object("0/sd/t", memory_budget_bytes=64 * 1024)
        assert torch.equal(t, sd["t"])


def test_get_state_dict_for_key():
    model = _Model()
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"model": model})
        sd = snapshot.get_state_dict_for_key("model")
        assert check_state_dict_eq(dict(model.state_dict()), dict(sd))


def test_restore_strict_false():
    """A snapshot saved from a smaller module restores into a bigger one
    with strict=False (missing keys keep their init values)."""
    small = torch.nn.Linear(4, 4)

    class Bigger(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.weight = torch.nn.Parameter(torch.zeros(4, 4))
            self.bias = torch.nn.Parameter(torch.zeros(4))
            self.extra = torch.nn.Parameter(torch.ones(3))

    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"m": small})
        big = Bigger()
        snapshot.restore({"m": big}, strict=False)
        assert torch.equal(big.weight.data, small.weight.data)
        assert torch.equal(big.bias.data, small.bias.data)
        assert torch.equal(big.extra.data, torch.ones(3))
        with pytest.raises(RuntimeError):
            snapshot.restore({"m": Bigger()}, strict=True)


def test_empty_app_state():
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {})
        assert snapshot.get_manifest() == {}
        snapshot.restore({})


def test_invalid_app_state_keys():
    with tmp_snapshot_path() as path:
        with pytest.raises(ValueError, match="invalid"):
            Snapshot.take(path, {"bad/key": StateDict(a=1)})
        with pytest.raises(TypeError):
            Snapshot.take(path, {"x": object()})


def test_snapshot_delete_fs():
    sd = StateDict(a=torch.rand(8))
    with tmp_snapshot_path() as path:
        snapshot = Snapshot.take(path, {"sd": sd})
        assert os.path.exists(path)
        snapshot.delete()
        assert not os.path.exists(path)
        with pytest.raises(RuntimeError):
            _ = Snapshot(path).metadata


def test_non_dict_state():
    """A stateful whose state_dict() returns a bare tensor (not a dict)."""

    class Weird:
        def __init__(self, t):
            self.t = t

        def state_dict(self):
            return self.t

        def load_state_dict(self, sd):
            self.t = sd

    w = Weird(torch.rand(7, 3))
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"w": w})
        w2 = Weird(torch.zeros(7, 3))
        snap.restore({"w": w2})
        assert torch.equal(w2.t, w.t)


def test_snapshot_relocatable():
    """All payload locations are relative: a moved snapshot directory
    restores identically."""
    import shutil

    sd = StateDict(a=torch.rand(32, 8), n=5)
    with tmp_snapshot_path() as path:
        Snapshot.take(path, {"sd": sd})
        moved = path + "_moved"
        shutil.move(path, moved)
        out = StateDict()
        Snapshot(moved).restore({"sd": out})
        assert torch.equal(out["a"], sd["a"])
        assert out["n"] == 5


def test_corrupted_metadata_message():
    with tmp_snapshot_path() as path:
        Snapshot.take(path, {"sd": StateDict(a=1)})
        with open(os.path.join(path, ".snapshot_metadata"), "w") as f:
            f.write("{not valid json or yaml: [")
        with pytest.raises(ValueError, match="corrupted"):
            _ = Snapshot(path).metadata


def test_fsync_mode(monkeypatch):
    monkeypatch.setenv("TSAMD_FSYNC", "1")
    sd = StateDict(a=torch.rand(64, 16))
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict()
        snap.restore({"sd": out})
        assert torch.equal(out["a"], sd["a"])


def test_custom_tensor_prepare_func():
    """Save-time transform: persist fp32 weights as bf16."""
    sd = StateDict(w=torch.rand(64, 32))
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(
            path,
            {"sd": sd},
            _custom_tensor_prepare_func=lambda p, t: t.to(torch.bfloat16),
        )
        assert snap.get_manifest()["0/sd/w"]["dtype"] == "bfloat16"
        out = StateDict(w=torch.zeros(64, 32))
        snap.restore({"sd": out})
        assert out["w"].dtype == torch.float32  # cast back on load
        assert torch.equal(out["w"], sd["w"].to(torch.bfloat16).float())


def test_tied_weights_written_once():
    """Tied parameters (same tensor under two state-dict keys) produce ONE
    payload; both paths restore."""

    class Tied(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.embed = torch.nn.Embedding(64, 16)
            self.head = torch.nn.Linear(16, 64, bias=False)
            self.head.weight = self.embed.weight  # tie

    m = Tied()
    with tmp_snapshot_path() as path:
        with __import__("torchsnapshot_amd").knobs.override_batching_disabled(True):
            snap = Snapshot.take(path, {"m": m})
        manifest = snap.get_manifest()
        e1 = manifest["0/m/embed.weight"]
        e2 = manifest["0/m/head.weight"]
        assert e1["location"] == e2["location"]
        # exactly one payload file for the tied weight
        payloads = []
        for root, _, names in os.walk(path):
            payloads += [os.path.join(root, n) for n in names if "weight" in n]
        assert len(payloads) == 1, payloads

        m2 = Tied()
        snap.restore({"m": m2})
        assert torch.equal(m2.embed.weight, m.embed.weight)
        assert torch.equal(m2.head.weight, m.head.weight)
        assert m2.head.weight.data_ptr() == m2.embed.weight.data_ptr()


def test_tied_weights_batched():
    class Tied(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.a = torch.nn.Linear(32, 32, bias=False)
            self.b = torch.nn.Linear(32, 32, bias=False)
            self.c = torch.nn.Linear(32, 32, bias=False)
            self.b.weight = self.a.weight

    m = Tied()
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"m": m})
        manifest = snap.get_manifest()
        assert (
            manifest["0/m/a.weight"]["location"]
            == manifest["0/m/b.weight"]["location"]
        )
        assert manifest["0/m/a.weight"].get("byte_range") == manifest[
            "0/m/b.weight"
        ].get("byte_range")
        m2 = Tied()
        snap.restore({"m": m2})
        assert torch.equal(m2.a.weight, m.a.weight)
        assert torch.equal(m2.c.weight, m.c.weight)


def test_parallel_segment_fs_io(monkeypatch):
    """Large files go through the concurrent-segment fs path; round-trip
    stays bit-exact (cold-read tuning, VERDICT round-1 item 4)."""
    import torch

    from torchsnapshot_amd import Snapshot, StateDict
    from torchsnapshot_amd.test_utils import tmp_snapshot_path

    monkeypatch.setenv("TSAMD_FS_PARALLEL_IO_MIN_BYTES", str(64 * 1024))
    monkeypatch.setenv("TSAMD_FS_IO_SEGMENT_BYTES", str(17 * 1024))  # odd size
    sd = StateDict(big=torch.rand(512, 257), small=torch.rand(3))
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict()
        snap.restore({"sd": out})
        assert torch.equal(out["big"], sd["big"])
        assert torch.equal(out["small"], sd["small"])


def test_parallel_segment_fs_io_fsync(monkeypatch):
    import torch

    from torchsnapshot_amd import Snapshot, StateDict
    from torchsnapshot_amd.test_utils import tmp_snapshot_path

    monkeypatch.setenv("TSAMD_FS_PARALLEL_IO_MIN_BYTES", str(64 * 1024))
    monkeypatch.setenv("TSAMD_FS_IO_SEGMENT_BYTES", str(32 * 1024))
    monkeypatch.setenv("TSAMD_FSYNC", "1")
    sd = StateDict(big=torch.rand(300, 300))
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict()
        snap.restore({"sd": out})
        assert torch.equal(out["big"], sd["big"])


def test_many_snapshots_loop_stability():
    """50 back-to-back take/async_take/restore cycles: no pipeline-thread,
    event-loop, or store-state leakage across snapshots."""
    import torch

    from torchsnapshot_amd import Snapshot, StateDict
    from torchsnapshot_amd.test_utils import tmp_snapshot_path

    sd = StateDict(w=torch.rand(64, 64), step=0)
    with tmp_snapshot_path() as path:
        for i in range(50):
            sd["step"] = i
            if i % 2:
                snap = Snapshot.async_take(path, {"sd": sd}).wait()
            else:
                snap = Snapshot.take(path, {"sd": sd})
        out = StateDict()
        snap.restore({"sd": out})
        assert out["step"] == 49
        assert torch.equal(out["w"], sd["w"])


def test_sparse_tensor_round_trip():
    """Sparse tensors route through torch_save (no flat storage) and
    survive save/restore — sync and async, with shadowing falling back
    gracefully."""
    import torch

    from torchsnapshot_amd import Snapshot, StateDict
    from torchsnapshot_amd.test_utils import tmp_snapshot_path

    sp = torch.sparse_coo_tensor([[0, 2], [1, 0]], [1.5, -2.0], (3, 3))
    sd = StateDict(dense=torch.rand(8), sp=sp)
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict()
        snap.restore({"sd": out})
        assert torch.equal(out["sp"].to_dense(), sp.to_dense())
        assert torch.equal(out["dense"], sd["dense"])

        pending = Snapshot.async_take(path + "2", {"sd": sd})
        assert not pending.sources_immutable  # sparse forces classic path
        snap2 = pending.wait()
        out2 = StateDict()
        snap2.restore({"sd": out2})
        assert torch.equal(out2["sp"].to_dense(), sp.to_dense())
