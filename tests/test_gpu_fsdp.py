"""FSDP adapter test on a single GPU (RCCL world of 1)."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs a ROCm GPU", allow_module_level=True)

import torch.distributed as dist  # noqa: E402

from torchsnapshot_amd import Snapshot  # noqa: E402
from torchsnapshot_amd.test_utils import tmp_snapshot_path  # noqa: E402
from torchsnapshot_amd.tricks import FSDPOptimizerAdapter  # noqa: E402


@pytest.fixture()
def dist_world1():
    created = False
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29532")
        dist.init_process_group(
            "nccl", rank=0, world_size=1, device_id=torch.device("cuda", 0)
        )
        created = True
    yield
    if created:
        dist.destroy_process_group()


def test_fsdp_model_and_optimizer_adapter(dist_world1):
    from torch.distributed.fsdp import FullyShardedDataParallel as FSDP

    torch.manual_seed(0)
    model = FSDP(torch.nn.Linear(64, 64).cuda())
    optim = torch.optim.Adam(model.parameters(), lr=1e-3)
    model(torch.rand(4, 64, device="cuda")).sum().backward()
    optim.step()

    with tmp_snapshot_path() as path:
        snap = Snapshot.take(
            path,
            {"model": model, "optim": FSDPOptimizerAdapter(model, optim)},
        )
        # perturb, then restore
        with torch.no_grad():
            for p in model.parameters():
                p.add_(1.0)
        before = [p.clone() for p in model.parameters()]
        snap.restore(
            {"model": model, "optim": FSDPOptimizerAdapter(model, optim)}
        )
        after = list(model.parameters())
        assert not all(torch.equal(b, a) for b, a in zip(before, after))
        # optimizer state survived the round trip
        osd = optim.state_dict()
        assert any("exp_avg" in str(v) for v in osd["state"].values()) or osd[
            "state"
        ]
