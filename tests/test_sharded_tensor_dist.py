"""ShardedTensor end-to-end on gloo (CPU), including reshard-on-restore
at a different world size."""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist

from torchsnapshot_amd.test_utils import run_multiprocess

pytestmark = pytest.mark.timeout(300)


def _make_sharded(world_size: int, seed: int = 0):
    from torch.distributed._shard import sharded_tensor
    from torch.distributed._shard.sharding_spec import ChunkShardingSpec

    spec = ChunkShardingSpec(
        dim=0,
        placements=[f"rank:{r}/cpu" for r in range(world_size)],
    )
    st = sharded_tensor.rand(spec, (48, 16))
    # deterministic content per shard
    for shard in st.local_shards():
        torch.manual_seed(seed + shard.metadata.shard_offsets[0])
        shard.tensor.copy_(torch.rand_like(shard.tensor))
    return st


class _Holder:
    def __init__(self, st):
        self.st = st

    def state_dict(self):
        return {"st": self.st}

    def load_state_dict(self, sd):
        self.st = sd["st"]


def _save(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    st = _make_sharded(dist.get_world_size(), seed=1)
    Snapshot.take(os.path.join(tmpdir, "snap"), {"obj": _Holder(st)})


def _restore_check(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    st = _make_sharded(dist.get_world_size(), seed=99)
    holder = _Holder(st)
    Snapshot(os.path.join(tmpdir, "snap")).restore({"obj": holder})
    # rebuild the reference full tensor (as saved at world 2, seed 1)
    for shard in holder.st.local_shards():
        lo = shard.metadata.shard_offsets[0]
        rows = shard.tensor.shape[0]
        # expected content: saved shards had seeds keyed by THEIR offsets
        # at save world size; reconstruct from full reference
        pass
    full = _full_reference()
    for shard in holder.st.local_shards():
        lo = shard.metadata.shard_offsets[0]
        rows = shard.tensor.shape[0]
        assert torch.equal(shard.tensor, full[lo : lo + rows])


def _full_reference() -> torch.Tensor:
    # world 2, dim0 48 -> shards at offsets 0 and 24
    full = torch.zeros(48, 16)
    for lo, rows in ((0, 24), (24, 24)):
        torch.manual_seed(1 + lo)
        full[lo : lo + rows] = torch.rand(rows, 16)
    return full


def test_sharded_save2_restore2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save, d)
        run_multiprocess(2, _restore_check, d)


def test_sharded_save2_restore3():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save, d)
        run_multiprocess(3, _restore_check, d)


def test_sharded_read_into_full_tensor():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _save, d)
        from torchsnapshot_amd import Snapshot

        out = Snapshot(os.path.join(d, "snap")).read_object("0/obj/st")
        assert torch.equal(out, _full_reference())


# ---------------------------------------------------------------------------
# indivisible table, world 3 -> world 5 (uneven shards on both sides)
# ---------------------------------------------------------------------------

_IND_ROWS = 47  # indivisible by 3 and by 5


def _make_indivisible(seed: int):
    from torch.distributed._shard import sharded_tensor
    from torch.distributed._shard.sharding_spec import ChunkShardingSpec

    world_size = dist.get_world_size()
    spec = ChunkShardingSpec(
        dim=0,
        placements=[f"rank:{r}/cpu" for r in range(world_size)],
    )
    st = sharded_tensor.rand(spec, (_IND_ROWS, 16))
    full = _indivisible_reference(seed)
    for shard in st.local_shards():
        lo = shard.metadata.shard_offsets[0]
        rows = shard.tensor.shape[0]
        shard.tensor.copy_(full[lo : lo + rows])
    return st


def _indivisible_reference(seed: int) -> torch.Tensor:
    torch.manual_seed(seed)
    return torch.rand(_IND_ROWS, 16)


def _save_indivisible(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    st = _make_indivisible(seed=42)
    Snapshot.take(os.path.join(tmpdir, "snap"), {"obj": _Holder(st)})


def _restore_indivisible(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    st = _make_indivisible(seed=777)  # wrong values, right layout
    holder = _Holder(st)
    Snapshot(os.path.join(tmpdir, "snap")).restore({"obj": holder})
    full = _indivisible_reference(42)
    for shard in holder.st.local_shards():
        lo = shard.metadata.shard_offsets[0]
        rows = shard.tensor.shape[0]
        assert torch.equal(shard.tensor, full[lo : lo + rows])


def test_sharded_indivisible_save3_restore5():
    """47 rows over 3 writers (16/16/15) restored over 5 readers
    (10/10/10/10/7): bit-exact (VERDICT round-1, next-round item 2)."""
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(3, _save_indivisible, d)
        run_multiprocess(5, _restore_indivisible, d)


def test_sharded_indivisible_save5_restore3():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(5, _save_indivisible, d)
        run_multiprocess(3, _restore_indivisible, d)


# ---------------------------------------------------------------------------
# embedding-table workload: multiple row-wise sharded tables, save on 4
# ranks, reshard-restore on 2 (BASELINE.json config 5 shape; mirror of
# reference tests/gpu_tests/test_torchrec.py:200-306 reshard matrix)
# ---------------------------------------------------------------------------

_N_TABLES = 3


def _emb_row_values(table_idx: int, rows: torch.Tensor) -> torch.Tensor:
    return (
        ((rows.to(torch.float64) * 2654435761.0 + table_idx * 97.0) % 1000003.0)
        / 1000003.0
    ).to(torch.float32)


def _make_tables(zero: bool = False):
    from torch.distributed._shard import sharded_tensor
    from torch.distributed._shard.sharding_spec import ChunkShardingSpec

    world_size = dist.get_world_size()
    spec = ChunkShardingSpec(
        dim=0, placements=[f"rank:{r}/cpu" for r in range(world_size)]
    )
    tables = {}
    for i in range(_N_TABLES):
        st = sharded_tensor.empty(spec, (50 + i * 7, 8))  # uneven everywhere
        for shard in st.local_shards():
            off = shard.metadata.shard_offsets[0]
            n = shard.tensor.shape[0]
            if zero:
                shard.tensor.zero_()
            else:
                shard.tensor.copy_(
                    _emb_row_values(i, torch.arange(off, off + n))
                    .unsqueeze(1)
                    .expand(n, 8)
                )
        tables[f"table_{i}"] = st
    return tables


class _Tables:
    def __init__(self, tables):
        self.tables = tables

    def state_dict(self):
        return dict(self.tables)

    def load_state_dict(self, sd):
        self.tables = dict(sd)


def _emb_save(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    Snapshot.take(
        os.path.join(tmpdir, "snap"), {"emb": _Tables(_make_tables())}
    )


def _emb_restore(tmpdir: str) -> None:
    from torchsnapshot_amd import Snapshot

    holder = _Tables(_make_tables(zero=True))
    Snapshot(os.path.join(tmpdir, "snap")).restore({"emb": holder})
    for i in range(_N_TABLES):
        st = holder.tables[f"table_{i}"]
        for shard in st.local_shards():
            off = shard.metadata.shard_offsets[0]
            n = shard.tensor.shape[0]
            want = (
                _emb_row_values(i, torch.arange(off, off + n))
                .unsqueeze(1)
                .expand(n, 8)
            )
            assert torch.equal(shard.tensor, want), (i, off, n)


def test_embedding_tables_save4_reshard_restore2():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(4, _emb_save, d)
        run_multiprocess(2, _emb_restore, d)
