"""Distributed CPU tests: DDP save/restore, replication, world-size
elasticity. Each test spawns gloo-connected subprocesses."""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist

from torchsnapshot_amd.test_utils import run_multiprocess

TIMEOUT = 120
pytestmark = pytest.mark.timeout(300)


class _Net(torch.nn.Module):
    def __init__(self):
        super().__init__()
        torch.manual_seed(42 + dist.get_rank() if dist.is_initialized() else 42)
        self.a = torch.nn.Linear(13, 29)
        self.b = torch.nn.Linear(29, 4)

    def forward(self, x):
        return self.b(torch.relu(self.a(x)))


def _ddp_save(path: str) -> None:
    from torch.nn.parallel import DistributedDataParallel as DDP

    from torchsnapshot_amd import Snapshot

    torch.manual_seed(100 + dist.get_rank())  # different init per rank
    net = _Net()
    ddp = DDP(net)  # DDP broadcasts rank 0's weights
    optim = torch.optim.SGD(ddp.parameters(), lr=0.1)
    ddp(torch.rand(4, 13)).sum().backward()
    optim.step()
    Snapshot.take(path, {"model": ddp, "optim": optim})

    # replication inferred from DDP: payload written exactly once
    files = []
    for root, _, names in os.walk(path):
        files.extend(os.path.join(root, n) for n in names)
    rep_files = [f for f in files if "/replicated/" in f or "/batched/" in f]
    assert rep_files, "expected replicated/batched payloads for DDP model"
    assert not any(f"/{r}/model" in f for f in files for r in (0, 1)), (
        "DDP model params must not be written per-rank"
    )


def _ddp_restore(path: str) -> None:
    from torch.nn.parallel import DistributedDataParallel as DDP

    from torchsnapshot_amd import Snapshot
    from torchsnapshot_amd.test_utils import check_state_dict_eq

    torch.manual_seed(7 + dist.get_rank())
    ddp = DDP(_Net())
    before = {k: v.clone() for k, v in ddp.state_dict().items()}
    snapshot = Snapshot(path)
    snapshot.restore({"model": ddp})
    after = ddp.state_dict()
    assert not check_state_dict_eq(before, after)
    # all ranks converge to the saved weights
    gathered = [None] * dist.get_world_size()
    dist.all_gather_object(
        gathered, {k: v.sum().item() for k, v in after.items()}
    )
    assert all(g == gathered[0] for g in gathered)


def _ddp_save_restore(tmpdir: str) -> None:
    path = os.path.join(tmpdir, "snap")
    _ddp_save(path)
    _ddp_restore(path)


def test_ddp_save_restore_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _ddp_save_restore, d)


def _save_world2(tmpdir: str) -> None:
    _ddp_save(os.path.join(tmpdir, "snap"))


def test_ddp_restore_upscaled_world():
    """Save at world 2, restore at world 3 (elasticity: new ranks borrow
    replicated entries)."""
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_world2, d)
        run_multiprocess(3, _restore_any_world, d)


def _restore_any_world(tmpdir: str) -> None:
    _ddp_restore(os.path.join(tmpdir, "snap"))


def test_ddp_restore_single_process():
    """Save at world 2, restore in ONE process without dist initialized,
    via the plain-module prefix adapter."""
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_world2, d)
        from torchsnapshot_amd import Snapshot
        from torchsnapshot_amd.tricks import StripDDPPrefixAdapter

        net = _Net()
        snapshot = Snapshot(os.path.join(d, "snap"))
        snapshot.restore({"model": StripDDPPrefixAdapter(net)})
        sd = snapshot.get_state_dict_for_key("model")
        for k, v in net.state_dict().items():
            assert torch.equal(v, sd["module." + k])


def _plain_save_ddp_restore(tmpdir: str) -> None:
    """Save a PLAIN module's snapshot, restore into a DDP wrapper via
    DDPWrappedAdapter."""
    from torch.nn.parallel import DistributedDataParallel as DDP

    from torchsnapshot_amd import Snapshot
    from torchsnapshot_amd.tricks import DDPWrappedAdapter

    path = os.path.join(tmpdir, "plain")
    if dist.get_rank() == 0:
        torch.manual_seed(55)
    torch.manual_seed(55)  # same weights everywhere for determinism
    plain = _Net()
    Snapshot.take(path, {"model": plain})

    torch.manual_seed(900 + dist.get_rank())
    ddp = DDP(_Net())
    Snapshot(path).restore({"model": DDPWrappedAdapter(ddp)})
    for (name, p), (_, expect) in zip(
        ddp.module.state_dict().items(), plain.state_dict().items()
    ):
        assert torch.equal(p, expect), name


def test_plain_save_restore_into_ddp():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _plain_save_ddp_restore, d)
