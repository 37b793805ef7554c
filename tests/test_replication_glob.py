"""Explicit replicated= glob handling: dedup, balance, cross-rank
agreement (reference tests/test_replication_glob.py pattern)."""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist

from torchsnapshot_amd.test_utils import run_multiprocess

pytestmark = pytest.mark.timeout(300)


def _save_with_globs(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot
    from torchsnapshot_amd.state_dict import StateDict

    import torchsnapshot_amd.knobs as knobs

    torch.manual_seed(0)  # identical on all ranks
    sd = StateDict(
        shared_a=torch.rand(256, 64),
        shared_b=torch.rand(128, 32),
        per_rank=torch.rand(10) + dist.get_rank(),
    )
    path = os.path.join(tmpdir, "snap")
    # batching off so the on-disk layout is directly inspectable
    with knobs.override_batching_disabled(True):
        Snapshot.take(path, {"app": sd}, replicated=["app/shared_*"])

    # replicated payloads exist exactly once; per-rank payloads per rank
    files = []
    for root, _, names in os.walk(path):
        files.extend(
            os.path.relpath(os.path.join(root, n), path) for n in names
        )
    rep = [f for f in files if f.startswith("replicated/")]
    assert rep, files
    for f in rep:
        assert files.count(f) == 1
    assert any(f.startswith("0/app/per_rank") for f in files)
    assert any(f.startswith("1/app/per_rank") for f in files)


def _restore_with_globs(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot
    from torchsnapshot_amd.state_dict import StateDict

    torch.manual_seed(0)
    expect_a = torch.rand(256, 64)
    expect_b = torch.rand(128, 32)
    sd = StateDict(
        shared_a=torch.zeros(256, 64),
        shared_b=torch.zeros(128, 32),
        per_rank=torch.zeros(10),
    )
    Snapshot(os.path.join(tmpdir, "snap")).restore({"app": sd})
    assert torch.equal(sd["shared_a"], expect_a)
    assert torch.equal(sd["shared_b"], expect_b)
    assert torch.allclose(sd["per_rank"] - dist.get_rank(), sd["per_rank"] - dist.get_rank())
    # per-rank value was saved per rank: rank r gets its own back
    assert (sd["per_rank"] >= dist.get_rank()).all()


def _save_and_restore(tmpdir: str) -> None:
    _save_with_globs(tmpdir)
    _restore_with_globs(tmpdir)


def test_replication_glob_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_and_restore, d)


def _save_mismatched_globs(tmpdir: str) -> None:
    """A glob requested on only one rank must be ignored (with a warning),
    not deadlock or diverge."""
    from torchsnapshot_amd import Snapshot
    from torchsnapshot_amd.state_dict import StateDict

    torch.manual_seed(0)
    sd = StateDict(x=torch.rand(16))
    replicated = ["app/**"] if dist.get_rank() == 0 else []
    path = os.path.join(tmpdir, "snap")
    Snapshot.take(path, {"app": sd}, replicated=replicated)
    # not replicated anywhere -> written per rank
    assert os.path.exists(os.path.join(path, "0", "app", "x"))
    assert os.path.exists(os.path.join(path, "1", "app", "x"))


def test_mismatched_globs_ignored():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_mismatched_globs, d)


def _save_chunked_replicated(tmpdir: str) -> None:
    """A replicated tensor bigger than the chunk size: chunks spread over
    ranks, manifest merges back on load."""
    import torchsnapshot_amd.knobs as knobs
    from torchsnapshot_amd import Snapshot
    from torchsnapshot_amd.state_dict import StateDict

    torch.manual_seed(7)
    big = torch.rand(1024, 128)  # 512 KB
    with knobs.override_max_chunk_size_bytes(64 * 1024):
        sd = StateDict(big=big)
        path = os.path.join(tmpdir, "snap")
        Snapshot.take(path, {"app": sd}, replicated=["**"])
    out = StateDict(big=torch.zeros(1024, 128))
    Snapshot(path).restore({"app": out})
    assert torch.equal(out["big"], big)
    # both ranks should have written some chunks (load balancing)
    chunk_dir = os.path.join(path, "replicated", "app")
    if dist.get_rank() == 0 and os.path.isdir(chunk_dir):
        names = os.listdir(chunk_dir)
        assert len(names) >= 2


def test_chunked_replicated_spread():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_chunked_replicated, d)


def _save_partitioner_disabled(tmpdir: str) -> None:
    """TSAMD_DISABLE_PARTITIONER: every rank writes replicated payloads
    (identical bytes to identical paths — safe); restore still works."""
    import torchsnapshot_amd.knobs as knobs
    from torchsnapshot_amd import Snapshot
    from torchsnapshot_amd.state_dict import StateDict

    torch.manual_seed(0)
    shared = torch.rand(64, 16)
    sd = StateDict(shared=shared.clone())
    path = os.path.join(tmpdir, "snap")
    with knobs.override_env("TSAMD_DISABLE_PARTITIONER", "1"):
        Snapshot.take(path, {"app": sd}, replicated=["**"])
    out = StateDict(shared=torch.zeros(64, 16))
    Snapshot(path).restore({"app": out})
    assert torch.equal(out["shared"], shared)


def test_partitioner_disabled_world2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save_partitioner_disabled, d)
