"""DDP replication inference (mirror of reference
tests/test_ddp_infer_replication.py): DDP modules are auto-marked
replicated, minus parameters_to_ignore."""

import pytest
import torch
import torch.distributed as dist

from torchsnapshot_amd.test_utils import run_multiprocess

pytestmark = pytest.mark.timeout(300)


def _model():
    return torch.nn.Sequential(torch.nn.Linear(4, 2), torch.nn.Linear(2, 1))


def _infer_no_ignore() -> None:
    from torch.nn.parallel import DistributedDataParallel as DDP

    from torchsnapshot_amd import Snapshot

    model = _model()
    inferred = Snapshot._infer_replicated(
        {"ddp": DDP(model), "nonddp": model}
    )
    assert sorted(inferred) == ["ddp/**"]


def _infer_with_ignore() -> None:
    from torch.nn.parallel import DistributedDataParallel as DDP

    from torchsnapshot_amd import Snapshot

    model = _model()
    DDP._set_params_and_buffers_to_ignore_for_model(
        model, ["module.0.bias", "module.0.weight"]
    )
    ddp_model = DDP(model)
    inferred = Snapshot._infer_replicated({"ddp": ddp_model, "nonddp": model})
    assert sorted(inferred) == ["ddp/module.1.bias", "ddp/module.1.weight"]


def _e2e_ignored_params_stay_per_rank(tmpdir: str) -> None:
    import os

    from torch.nn.parallel import DistributedDataParallel as DDP

    from torchsnapshot_amd import Snapshot

    torch.manual_seed(0)
    model = _model()
    DDP._set_params_and_buffers_to_ignore_for_model(
        model, ["module.0.bias", "module.0.weight"]
    )
    ddp_model = DDP(model)
    # the ignored params are NOT synced by DDP: give them rank-specific
    # values, which the snapshot must preserve per rank
    rank = dist.get_rank()
    with torch.no_grad():
        model[0].weight.fill_(float(rank))
        model[0].bias.fill_(float(rank))
    path = os.path.join(tmpdir, "snap")
    Snapshot.take(path, {"ddp": ddp_model})

    with torch.no_grad():
        model[0].weight.zero_()
        model[0].bias.zero_()
        model[1].weight.zero_()
    Snapshot(path).restore({"ddp": ddp_model})
    assert torch.equal(
        model[0].weight, torch.full_like(model[0].weight, float(rank))
    )
    assert torch.equal(
        model[0].bias, torch.full_like(model[0].bias, float(rank))
    )


def test_infer_replicated_world2():
    import tempfile

    run_multiprocess(2, _infer_no_ignore)
    run_multiprocess(2, _infer_with_ignore)
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _e2e_ignored_params_stay_per_rank, d)
