"""Kitchen-sink integration: one snapshot holding replicated tensors,
per-rank state, a ShardedTensor, a DTensor, primitives and an object —
saved at world 4, restored at world 4 and world 2, plus single-process
random access."""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist

from torchsnapshot_amd.test_utils import run_multiprocess

pytestmark = pytest.mark.timeout(600)


def _make_state(world_size: int):
    from torch.distributed._shard import sharded_tensor
    from torch.distributed._shard.sharding_spec import ChunkShardingSpec
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Shard

    torch.manual_seed(3)
    shared = torch.rand(64, 16)

    spec = ChunkShardingSpec(
        dim=0, placements=[f"rank:{r}/cpu" for r in range(world_size)]
    )
    st = sharded_tensor.zeros(spec, (32, 8))
    for shard in st.local_shards():
        torch.manual_seed(100 + shard.metadata.shard_offsets[0])
        shard.tensor.copy_(torch.rand_like(shard.tensor))

    mesh = init_device_mesh("cpu", (world_size,))
    torch.manual_seed(7)
    dt_full = torch.rand(16, 4)
    dt = distribute_tensor(dt_full, mesh, [Shard(0)])

    rank = dist.get_rank()
    return {
        "shared": shared,
        "st": st,
        "dt": dt,
        "mine": torch.full((4,), float(rank)),
        "step": 123,
        "name": "kitchen",
        "blob": (1, 2, 3),
    }


class _Holder:
    def __init__(self, sd):
        self._sd = sd

    def state_dict(self):
        return self._sd

    def load_state_dict(self, sd):
        self._sd = sd


def _save(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    state = _make_state(dist.get_world_size())
    Snapshot.take(
        os.path.join(tmpdir, "snap"),
        {"app": _Holder(state)},
        replicated=["app/shared"],
    )


def _restore(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    ws = dist.get_world_size()
    rank = dist.get_rank()
    state = _make_state(ws)
    # zero the in-place targets
    state["shared"] = torch.zeros(64, 16)
    holder = _Holder(state)
    Snapshot(os.path.join(tmpdir, "snap")).restore({"app": holder})
    out = holder._sd

    torch.manual_seed(3)
    assert torch.equal(out["shared"], torch.rand(64, 16))
    torch.manual_seed(7)
    dt_full = torch.rand(16, 4)
    assert torch.equal(out["dt"].full_tensor(), dt_full)
    for shard in out["st"].local_shards():
        lo = shard.metadata.shard_offsets[0]
        rows = shard.tensor.shape[0]
        expect = _st_reference(saved_world=4)[lo : lo + rows]
        assert torch.equal(shard.tensor, expect)
    if rank < 4:
        # per-rank value survives for ranks that existed at save time
        assert torch.equal(out["mine"], torch.full((4,), float(rank)))
    assert out["step"] == 123
    assert out["name"] == "kitchen"
    assert out["blob"] == (1, 2, 3)


def _st_reference(saved_world: int) -> torch.Tensor:
    full = torch.zeros(32, 8)
    rows = 32 // saved_world
    for i in range(saved_world):
        lo = i * rows
        torch.manual_seed(100 + lo)
        full[lo : lo + rows] = torch.rand(rows, 8)
    return full


def test_kitchen_sink_world4():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(4, _save, d)
        run_multiprocess(4, _restore, d)
        # world-size change: restore at 2
        run_multiprocess(2, _restore, d)
        # single-process random access against the multi-rank snapshot
        from torchsnapshot_amd import Snapshot

        snap = Snapshot(os.path.join(d, "snap"))
        torch.manual_seed(3)
        assert torch.equal(snap.read_object("0/app/shared"), torch.rand(64, 16))
        assert torch.equal(
            snap.read_object("2/app/mine"), torch.full((4,), 2.0)
        )
        assert torch.equal(snap.read_object("1/app/st"), _st_reference(4))
        assert snap.read_object("3/app/step") == 123
