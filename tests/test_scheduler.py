"""Scheduler edge cases: tiny budgets (escape hatch), many requests,
error aggregation."""

import asyncio
import os
import tempfile

import pytest
import torch

from torchsnapshot_amd import Snapshot, StateDict, knobs
from torchsnapshot_amd.io_types import (
    BufferStager,
    StageContext,
    WriteReq,
)
from torchsnapshot_amd.scheduler import (
    execute_write_reqs,
    sync_execute_write_reqs,
)
from torchsnapshot_amd.storage.fs import FSStoragePlugin

pytestmark = pytest.mark.timeout(120)


class _BytesStager(BufferStager):
    def __init__(self, payload: bytes, fail: bool = False):
        self.payload = payload
        self.fail = fail

    def get_staging_cost_bytes(self) -> int:
        return len(self.payload)

    async def stage_buffer(self, ctx: StageContext):
        if self.fail:
            raise RuntimeError("stager failure")
        return self.payload


def test_budget_smaller_than_any_item_still_completes():
    """Every request is bigger than the whole budget: the empty-pipeline
    escape hatch must serialize them rather than deadlock."""
    with tempfile.TemporaryDirectory() as d:
        storage = FSStoragePlugin(d)
        reqs = [
            WriteReq(path=f"f{i}", stager=_BytesStager(bytes([i]) * 1000))
            for i in range(8)
        ]
        stats = sync_execute_write_reqs(
            reqs, storage, memory_budget_bytes=10, rank=0
        )
        assert stats.done_reqs == 8
        for i in range(8):
            assert os.path.getsize(os.path.join(d, f"f{i}")) == 1000
        storage.sync_close()


def test_many_small_requests():
    with tempfile.TemporaryDirectory() as d:
        storage = FSStoragePlugin(d)
        reqs = [
            WriteReq(path=f"n/{i}", stager=_BytesStager(b"x" * 10))
            for i in range(500)
        ]
        stats = sync_execute_write_reqs(
            reqs, storage, memory_budget_bytes=1 << 20, rank=0
        )
        assert stats.done_reqs == 500
        storage.sync_close()


def test_one_failing_stager_fails_pipeline_but_releases_all():
    with tempfile.TemporaryDirectory() as d:
        storage = FSStoragePlugin(d)
        reqs = [
            WriteReq(path=f"g{i}", stager=_BytesStager(b"y" * 100))
            for i in range(5)
        ] + [WriteReq(path="bad", stager=_BytesStager(b"", fail=True))]
        pending = execute_write_reqs(
            reqs, storage, memory_budget_bytes=1 << 20, rank=0
        )
        with pytest.raises(RuntimeError, match="stager failure"):
            pending.complete()
        storage.sync_close()


def test_tiny_memory_budget_e2e(monkeypatch):
    """A snapshot under an absurdly small memory budget still succeeds."""
    monkeypatch.setenv("TSAMD_PER_RANK_MEMORY_BUDGET_BYTES", "4096")
    sd = StateDict(a=torch.rand(64, 64), b=torch.rand(128), c=torch.rand(3, 3))
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "s")
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict()
        snap.restore({"sd": out})
        assert torch.equal(out["a"], sd["a"])
