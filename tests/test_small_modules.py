"""Coverage for the small cross-cutting modules: memoryview stream,
events, glob matching, knobs overrides, rss profiler, pg_wrapper no-dist
behavior."""

import io
import time
from collections import deque

import pytest
import torch

from torchsnapshot_amd import knobs
from torchsnapshot_amd.event import (
    Event,
    log_event,
    register_event_handler,
    unregister_event_handler,
)
from torchsnapshot_amd.memoryview_stream import MemoryviewStream
from torchsnapshot_amd.pg_wrapper import PGWrapper
from torchsnapshot_amd.rss_profiler import max_rss_delta_mb, measure_rss_deltas
from torchsnapshot_amd.snapshot import glob_match


def test_memoryview_stream():
    data = bytes(range(256))
    s = MemoryviewStream(memoryview(data))
    assert s.readable() and s.seekable()
    assert s.read(10) == data[:10]
    assert s.tell() == 10
    s.seek(100)
    assert s.read(5) == data[100:105]
    s.seek(-6, io.SEEK_END)
    assert s.read() == data[-6:]
    s.seek(0)
    buf = bytearray(300)
    assert s.readinto(buf) == 256
    assert bytes(buf[:256]) == data
    s.close()
    with pytest.raises(ValueError):
        s.read(1)


def test_glob_match():
    assert glob_match("model/lin.weight", "model/**")
    assert glob_match("model/a/b/c", "**")
    assert glob_match("model/lin.weight", "model/*.weight")
    assert not glob_match("model/sub/lin.weight", "model/*.weight")
    assert glob_match("model/sub/lin.weight", "model/**/*.weight")
    assert glob_match("x", "**")
    assert not glob_match("model/x", "optim/**")
    assert glob_match("a/b", "a/b")


def test_event_handlers():
    events = []

    class Handler:
        def handle_event(self, event: Event) -> None:
            events.append(event)

    h = Handler()
    register_event_handler(h)
    try:
        log_event(Event("test_event", {"k": 1}))
    finally:
        unregister_event_handler(h)
    assert events and events[0].name == "test_event"
    log_event(Event("after", {}))
    assert len(events) == 1  # unregistered


def test_events_emitted_by_take(tmp_path):
    from torchsnapshot_amd import Snapshot, StateDict

    events = []

    class Handler:
        def handle_event(self, event: Event) -> None:
            events.append(event.name)

    h = Handler()
    register_event_handler(h)
    try:
        snap = Snapshot.take(str(tmp_path / "s"), {"sd": StateDict(a=1)})
        snap.restore({"sd": StateDict()})
    finally:
        unregister_event_handler(h)
    assert "take_start" in events and "take_end" in events
    assert "restore_start" in events and "restore_end" in events


def test_knobs_overrides():
    with knobs.override_max_chunk_size_bytes(123):
        assert knobs.get_max_chunk_size_bytes() == 123
    assert knobs.get_max_chunk_size_bytes() != 123
    with knobs.override_batching_disabled(True):
        assert knobs.is_batching_disabled()
    assert not knobs.is_batching_disabled()
    with knobs.override_max_io_concurrency(3):
        assert knobs.get_max_io_concurrency() == 3


def test_rss_profiler():
    deltas = deque()
    with measure_rss_deltas(deltas, interval_s=0.01):
        blob = bytearray(64 * 1024 * 1024)
        blob[::4096] = b"x" * len(blob[::4096])
        time.sleep(0.1)
    assert len(deltas) > 0
    assert max_rss_delta_mb(deltas) >= 0


def test_pg_wrapper_no_dist():
    pgw = PGWrapper(None)
    assert pgw.get_rank() == 0
    assert pgw.get_world_size() == 1
    pgw.barrier()
    out = [None]
    pgw.all_gather_object(out, "x")
    assert out == ["x"]
    lst = ["payload"]
    pgw.broadcast_object_list(lst)
    assert lst == ["payload"]
    res = [None]
    pgw.scatter_object_list(res, ["only"])
    assert res == ["only"]


def test_uvm_fallbacks_on_cpu():
    from torchsnapshot_amd.uvm_tensor import is_uvm_tensor, uvm_to_cpu

    t = torch.rand(4)
    assert not is_uvm_tensor(t)
    assert uvm_to_cpu(t) is t


def test_cli_inspect_and_cat(tmp_path, capsys):
    import torch

    from torchsnapshot_amd import Snapshot, StateDict
    from torchsnapshot_amd.__main__ import main

    p = str(tmp_path / "snap")
    Snapshot.take(p, {"sd": StateDict(w=torch.rand(16, 4), n=9)})
    assert main(["inspect", p]) == 0
    out = capsys.readouterr().out
    assert "0/sd/w" in out and "tensor" in out
    assert main(["cat", p, "0/sd/n"]) == 0
    assert capsys.readouterr().out.strip() == "9"


# ---------------------------------------------------------------------------
# equality oracles (mirror of reference tests/test_test_utils.py)
# ---------------------------------------------------------------------------


def test_check_state_dict_eq_plain():
    import torch

    from torchsnapshot_amd.test_utils import (
        assert_state_dict_eq,
        check_state_dict_eq,
    )

    a = {"x": torch.ones(4), "n": 3, "nested": {"y": torch.zeros(2)}}
    b = {"x": torch.ones(4), "n": 3, "nested": {"y": torch.zeros(2)}}
    assert check_state_dict_eq(a, b)
    b["x"] = torch.zeros(4)
    assert not check_state_dict_eq(a, b)
    import unittest

    with pytest.raises(AssertionError):
        assert_state_dict_eq(unittest.TestCase(), a, b)


def test_check_state_dict_eq_dtype_and_shape_mismatch():
    import torch

    from torchsnapshot_amd.test_utils import check_state_dict_eq

    assert not check_state_dict_eq(
        {"x": torch.ones(4)}, {"x": torch.ones(4, dtype=torch.float64)}
    )
    assert not check_state_dict_eq({"x": torch.ones(4)}, {"x": torch.ones(5)})
    assert not check_state_dict_eq({"x": torch.ones(4)}, {})


def test_rand_tensor_dtypes():
    import torch

    from torchsnapshot_amd.test_utils import rand_tensor

    for dtype in (
        torch.float32,
        torch.bfloat16,
        torch.int64,
        torch.bool,
        torch.complex64,
    ):
        t = rand_tensor((4, 4), dtype)
        assert t.dtype == dtype and t.shape == (4, 4)
    q = rand_tensor((8,), torch.qint8)
    assert q.is_quantized


# ---------------------------------------------------------------------------
# third-party storage plugin discovery (entry points)
# ---------------------------------------------------------------------------


class _RamPlugin:
    """Minimal in-memory StoragePlugin registered via a fake entry point."""

    store = {}

    def __init__(self, root, storage_options=None):
        self.root = root

    async def write(self, write_io):
        self.store[f"{self.root}/{write_io.path}"] = bytes(
            memoryview(write_io.buf)
        )

    async def read(self, read_io):
        data = self.store[f"{self.root}/{read_io.path}"]
        if read_io.byte_range is not None:
            s, e = read_io.byte_range
            data = data[s:e]
        read_io.buf = bytearray(data)

    async def delete(self, path):
        self.store.pop(f"{self.root}/{path}", None)

    async def delete_dir(self, path):
        prefix = f"{self.root}/{path}"
        for k in [k for k in self.store if k.startswith(prefix)]:
            del self.store[k]


def test_third_party_storage_plugin_entry_point(monkeypatch):
    """A `tsamd_storage_plugins` entry point resolves custom protocols
    (parity with reference storage_plugin.py:55-67) and a full snapshot
    round-trips through it."""
    import torch

    import torchsnapshot_amd.storage as storage_mod
    from torchsnapshot_amd import Snapshot, StateDict
    from torchsnapshot_amd.io_types import StoragePlugin

    class _EP:
        name = "ram"

        @staticmethod
        def load():
            return _RamPlugin

    def fake_entry_points(group=None):
        assert group == "tsamd_storage_plugins"
        return [_EP]

    monkeypatch.setattr(storage_mod, "entry_points", fake_entry_points)
    # make sure the ABC sync wrappers exist on the minimal plugin
    for name in ("sync_write", "sync_read", "sync_close", "close",
                 "close_for_loop"):
        assert hasattr(StoragePlugin, name)
    _RamPlugin.sync_write = StoragePlugin.sync_write
    _RamPlugin.sync_read = StoragePlugin.sync_read
    _RamPlugin.sync_close = StoragePlugin.sync_close
    _RamPlugin.close = StoragePlugin.close
    _RamPlugin.close_for_loop = StoragePlugin.close_for_loop

    sd = StateDict(w=torch.rand(32, 8), n=5)
    snap = Snapshot.take("ram://bucket/ckpt", {"sd": sd})
    out = StateDict()
    snap.restore({"sd": out})
    assert torch.equal(out["w"], sd["w"]) and out["n"] == 5
