"""DTensor end-to-end on gloo/CPU meshes: sharded, replicated, and
2-D sharded+replicated (HSDP-style) layouts."""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist

from torchsnapshot_amd.test_utils import run_multiprocess

pytestmark = pytest.mark.timeout(300)


def _full(seed: int = 5) -> torch.Tensor:
    torch.manual_seed(seed)
    return torch.rand(48, 8)


class _Holder:
    def __init__(self, dt):
        self.dt = dt

    def state_dict(self):
        return {"dt": self.dt}

    def load_state_dict(self, sd):
        self.dt = sd["dt"]


def _save_sharded(tmpdir: str) -> None:
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Shard

    from torchsnapshot_amd import Snapshot

    mesh = init_device_mesh("cpu", (dist.get_world_size(),))
    dt = distribute_tensor(_full(), mesh, [Shard(0)])
    Snapshot.take(os.path.join(tmpdir, "snap"), {"obj": _Holder(dt)})


def _restore_sharded(tmpdir: str) -> None:
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Shard

    from torchsnapshot_amd import Snapshot

    mesh = init_device_mesh("cpu", (dist.get_world_size(),))
    dt = distribute_tensor(torch.zeros(48, 8), mesh, [Shard(0)])
    holder = _Holder(dt)
    Snapshot(os.path.join(tmpdir, "snap")).restore({"obj": holder})
    assert torch.equal(holder.dt.full_tensor(), _full())


def test_dtensor_shard_save_restore_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_sharded, d)
        run_multiprocess(2, _restore_sharded, d)


def test_dtensor_reshard_world2_to_world4():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_sharded, d)
        run_multiprocess(4, _restore_sharded, d)


def _save_replicated(tmpdir: str) -> None:
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Replicate

    from torchsnapshot_amd import Snapshot

    mesh = init_device_mesh("cpu", (dist.get_world_size(),))
    dt = distribute_tensor(_full(), mesh, [Replicate()])
    Snapshot.take(os.path.join(tmpdir, "snap"), {"obj": _Holder(dt)})
    # replicated DTensor payload written exactly once
    files = []
    for root, _, names in os.walk(tmpdir):
        files.extend(os.path.join(root, n) for n in names)
    payloads = [f for f in files if "replicated" in f]
    assert len(payloads) >= 1


def _restore_replicated(tmpdir: str) -> None:
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Replicate

    from torchsnapshot_amd import Snapshot

    mesh = init_device_mesh("cpu", (dist.get_world_size(),))
    dt = distribute_tensor(torch.zeros(48, 8), mesh, [Replicate()])
    holder = _Holder(dt)
    Snapshot(os.path.join(tmpdir, "snap")).restore({"obj": holder})
    assert torch.equal(holder.dt.full_tensor(), _full())


def test_dtensor_replicated_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_replicated, d)
        run_multiprocess(2, _restore_replicated, d)


def _save_hsdp(tmpdir: str) -> None:
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Replicate, Shard

    from torchsnapshot_amd import Snapshot

    mesh = init_device_mesh("cpu", (2, 2))
    dt = distribute_tensor(_full(), mesh, [Replicate(), Shard(0)])
    Snapshot.take(os.path.join(tmpdir, "snap"), {"obj": _Holder(dt)})
    # 2 shards, each replicated over 2 ranks: exactly 2 payload files
    files = []
    for root, _, names in os.walk(tmpdir):
        for n in names:
            if "replicated_sharded" in os.path.join(root, n):
                files.append(n)
    assert len(files) == 2, files


def _restore_hsdp(tmpdir: str) -> None:
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Replicate, Shard

    from torchsnapshot_amd import Snapshot

    mesh = init_device_mesh("cpu", (2, 2))
    dt = distribute_tensor(torch.zeros(48, 8), mesh, [Replicate(), Shard(0)])
    holder = _Holder(dt)
    Snapshot(os.path.join(tmpdir, "snap")).restore({"obj": holder})
    assert torch.equal(holder.dt.full_tensor(), _full())


def test_dtensor_hsdp_world4():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(4, _save_hsdp, d)
        run_multiprocess(4, _restore_hsdp, d)


def _save_shard_for_reading(tmpdir: str) -> None:
    _save_sharded(tmpdir)


def test_dtensor_read_object_full():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_shard_for_reading, d)
        from torchsnapshot_amd import Snapshot

        out = Snapshot(os.path.join(d, "snap")).read_object("0/obj/dt")
        assert torch.equal(out, _full())


def _save_replicated_with_glob(tmpdir: str) -> None:
    """A fully-replicated DTensor under replicated=['**'] must not lose
    its payload to partitioner reassignment (the preparer already chose
    its writer)."""
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Replicate

    from torchsnapshot_amd import Snapshot
    from torchsnapshot_amd.state_dict import StateDict

    mesh = init_device_mesh("cpu", (dist.get_world_size(),))
    dt = distribute_tensor(_full(), mesh, [Replicate()])
    torch.manual_seed(11)
    plain = torch.rand(32, 4)
    sd = StateDict(dt=dt, plain=plain)
    path = os.path.join(tmpdir, "snap")
    Snapshot.take(path, {"app": sd}, replicated=["**"])

    out = StateDict(
        dt=distribute_tensor(torch.zeros(48, 8), mesh, [Replicate()]),
        plain=torch.zeros(32, 4),
    )
    Snapshot(path).restore({"app": out})
    assert torch.equal(out["dt"].full_tensor(), _full())
    assert torch.equal(out["plain"], plain)


def test_replicated_dtensor_with_glob_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_replicated_with_glob, d)


# ---------------------------------------------------------------------------
# uneven (indivisible) DTensor shards: world 3 save, world 5 restore
# ---------------------------------------------------------------------------


def _uneven_full(seed: int = 21) -> torch.Tensor:
    torch.manual_seed(seed)
    return torch.rand(47, 8)  # 47 rows: uneven at world 3 and 5


def _save_uneven(tmpdir: str) -> None:
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Shard

    from torchsnapshot_amd import Snapshot

    mesh = init_device_mesh("cpu", (dist.get_world_size(),))
    dt = distribute_tensor(_uneven_full(), mesh, [Shard(0)])
    Snapshot.take(os.path.join(tmpdir, "snap"), {"obj": _Holder(dt)})


def _restore_uneven(tmpdir: str) -> None:
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Shard

    from torchsnapshot_amd import Snapshot

    mesh = init_device_mesh("cpu", (dist.get_world_size(),))
    dt = distribute_tensor(torch.zeros(47, 8), mesh, [Shard(0)])
    holder = _Holder(dt)
    Snapshot(os.path.join(tmpdir, "snap")).restore({"obj": holder})
    assert torch.equal(holder.dt.full_tensor(), _uneven_full())


def test_dtensor_uneven_save3_restore5():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(3, _save_uneven, d)
        run_multiprocess(5, _restore_uneven, d)


def _async_shadow_dtensor(tmpdir: str) -> None:
    import os as _os

    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import distribute_tensor
    from torch.distributed.tensor.placement_types import Shard

    from torchsnapshot_amd import Snapshot

    _os.environ["TSAMD_ASYNC_SHADOW"] = "1"
    try:
        mesh = init_device_mesh("cpu", (dist.get_world_size(),))
        dt = distribute_tensor(_full(31), mesh, [Shard(0)])
        holder = _Holder(dt)
        path = _os.path.join(tmpdir, "snap")
        pending = Snapshot.async_take(path, {"obj": holder})
        assert pending.sources_immutable
        # mutate the local shard immediately — the snapshot must not see it
        with torch.no_grad():
            dt.to_local().zero_()
        pending.wait()

        out = _Holder(distribute_tensor(torch.zeros(48, 8), mesh, [Shard(0)]))
        Snapshot(path).restore({"obj": out})
        assert torch.equal(out.dt.full_tensor(), _full(31))
    finally:
        del _os.environ["TSAMD_ASYNC_SHADOW"]


def test_async_shadow_dtensor_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _async_shadow_dtensor, d)
