"""Regressions from the round-1 advisor review:

1. Tied-weight dedup + replicated partitioner + batching: the dropped
   rank's manifest must not keep a tied alias pointing at a standalone
   path that the writer's batcher relocated away.
2. Replicated/HSDP DTensor whose local shard subdivides into multiple
   pieces: entries for pieces written by OTHER replica-set ranks must not
   shadow the writer's (batched/relocated) entries on load.
3. LinearBarrier keys must not leak across snapshots to the same path
   (second async_take to the same path, and a snapshot after a failed one).
"""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist

from torchsnapshot_amd.test_utils import run_multiprocess

pytestmark = pytest.mark.timeout(300)


class _Holder:
    def __init__(self, obj):
        self.obj = obj

    def state_dict(self):
        return {"t": self.obj}

    def load_state_dict(self, sd):
        self.obj = sd["t"]


# ---------------------------------------------------------------------------
# 1. tied weights x replicated partitioner x batching
# ---------------------------------------------------------------------------


class _TiedState:
    """Two logical paths aliasing one tensor + a per-rank filler that
    skews the partitioner so the replicated write lands on rank 1."""

    def __init__(self, shared: torch.Tensor, filler: torch.Tensor):
        self.shared = shared
        self.filler = filler

    def state_dict(self):
        return {"a": self.shared, "b": self.shared, "filler": self.filler}

    def load_state_dict(self, sd):
        self.shared = sd["a"]
        self.filler = sd["filler"]


def _tied_state() -> _TiedState:
    torch.manual_seed(7)
    shared = torch.rand(1024)
    rank = dist.get_rank()
    # rank 0 carries a big non-replicated write so the greedy partitioner
    # assigns the (replicated) shared tensor to rank 1
    filler = torch.rand(65536) if rank == 0 else torch.rand(16)
    return _TiedState(shared, filler)


def _tied_save_restore(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    path = os.path.join(tmpdir, "snap")
    state = _tied_state()
    Snapshot.take(path, {"m": state}, replicated=["m/a", "m/b"])

    target = _tied_state()
    target.shared = torch.zeros(1024)
    target.filler = torch.zeros_like(target.filler)
    Snapshot(path).restore({"m": target})
    torch.manual_seed(7)
    assert torch.equal(target.shared, torch.rand(1024))


def test_tied_replicated_batched_restore_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _tied_save_restore, d)


# ---------------------------------------------------------------------------
# 2. replicated DTensor with multi-piece shards + batching
# ---------------------------------------------------------------------------


def _full_dt(seed: int = 3) -> torch.Tensor:
    torch.manual_seed(seed)
    return torch.rand(64, 32)


def _repl_dtensor_save_restore(tmpdir: str) -> None:
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Replicate

    from torchsnapshot_amd import Snapshot

    # force each rank's (replicated) shard to subdivide into 4 pieces so
    # the round-robin writer choice splits them across the replica set AND
    # each rank holds >=2 write reqs (so the batcher actually slabs +
    # relocates its entries)
    os.environ["TSAMD_MAX_SHARD_SIZE_BYTES"] = "2048"
    try:
        path = os.path.join(tmpdir, "snap")
        mesh = init_device_mesh("cpu", (dist.get_world_size(),))
        dt = distribute_tensor(_full_dt(), mesh, [Replicate()])
        Snapshot.take(path, {"m": _Holder(dt)})

        dt2 = distribute_tensor(torch.zeros(64, 32), mesh, [Replicate()])
        holder = _Holder(dt2)
        Snapshot(path).restore({"m": holder})
        assert torch.equal(holder.obj.full_tensor(), _full_dt())
    finally:
        del os.environ["TSAMD_MAX_SHARD_SIZE_BYTES"]


def test_replicated_dtensor_batched_pieces_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _repl_dtensor_save_restore, d)


# ---------------------------------------------------------------------------
# 3. LinearBarrier key reuse across snapshots to the same path
# ---------------------------------------------------------------------------


def _async_take_same_path_twice(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    path = os.path.join(tmpdir, "snap")
    torch.manual_seed(11)
    state = _Holder(torch.rand(256))
    for i in range(3):
        state.obj = torch.full((256,), float(i))
        pending = Snapshot.async_take(path, {"m": state})
        snap = pending.wait()
    target = _Holder(torch.zeros(256))
    snap.restore({"m": target})
    assert torch.equal(target.obj, torch.full((256,), 2.0))


def test_async_take_same_path_reuse_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _async_take_same_path_twice, d)


# ---------------------------------------------------------------------------
# shadow clones x replicated partitioner x tied aliases (all round-2
# features composed, async, world 2)
# ---------------------------------------------------------------------------


def _tied_async_shadow(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    os.environ["TSAMD_ASYNC_SHADOW"] = "1"
    try:
        path = os.path.join(tmpdir, "snap")
        state = _tied_state()
        pending = Snapshot.async_take(
            path, {"m": state}, replicated=["m/a", "m/b"]
        )
        assert pending.sources_immutable
        # mutate everything immediately
        with torch.no_grad():
            state.shared.zero_()
            state.filler.zero_()
        pending.wait()

        target = _tied_state()
        target.shared = torch.zeros(1024)
        target.filler = torch.zeros_like(target.filler)
        Snapshot(path).restore({"m": target})
        torch.manual_seed(7)
        assert torch.equal(target.shared, torch.rand(1024))
    finally:
        del os.environ["TSAMD_ASYNC_SHADOW"]


def test_tied_replicated_async_shadow_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _tied_async_shadow, d)
