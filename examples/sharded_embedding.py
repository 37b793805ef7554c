#!/usr/bin/env python3
"""Sharded-embedding example (the torchrec DMP workload shape, mirror of
reference examples/torchrec without the torchrec dependency): row-wise
sharded embedding tables saved from N ranks and reshard-restored at any
world size.

Launch: python -m torch.distributed.run --nproc-per-node 2 \
            --master-addr 127.0.0.1 examples/sharded_embedding.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import tempfile

import torch
import torch.distributed as dist
from torch.distributed._shard import sharded_tensor
from torch.distributed._shard.sharding_spec import ChunkShardingSpec

from torchsnapshot_amd import Snapshot


class EmbeddingBag:
    """Stateful holding row-wise sharded tables (what torchrec's
    DistributedModelParallel produces for EmbeddingBagCollection)."""

    def __init__(self, tables):
        self.tables = tables

    def state_dict(self):
        return dict(self.tables)

    def load_state_dict(self, sd):
        self.tables = dict(sd)


def make_tables(world_size: int, zero: bool = False):
    spec = ChunkShardingSpec(
        dim=0, placements=[f"rank:{r}/cpu" for r in range(world_size)]
    )
    tables = {}
    for i, rows in enumerate((1000, 500, 250)):
        st = sharded_tensor.empty(spec, (rows, 64))
        for shard in st.local_shards():
            if zero:
                shard.tensor.zero_()
            else:
                off = shard.metadata.shard_offsets[0]
                n = shard.tensor.shape[0]
                shard.tensor.copy_(
                    torch.arange(off, off + n, dtype=torch.float32)
                    .unsqueeze(1)
                    .expand(n, 64)
                    + i
                )
        tables[f"table_{i}"] = st
    return tables


def main() -> None:
    dist.init_process_group(backend="gloo")
    rank = dist.get_rank()
    world_size = dist.get_world_size()

    tmp = tempfile.mkdtemp() if rank == 0 else None
    holder = [tmp]
    dist.broadcast_object_list(holder, src=0)
    path = os.path.join(holder[0], "snap")

    emb = EmbeddingBag(make_tables(world_size))
    Snapshot.take(path, {"emb": emb})
    if rank == 0:
        print(f"saved {len(emb.tables)} sharded tables from {world_size} ranks")

    # reshard-restore: same world size here, but any world size works —
    # each persisted shard is scattered into whatever local shards exist
    target = EmbeddingBag(make_tables(world_size, zero=True))
    Snapshot(path).restore({"emb": target})
    for i, rows in enumerate((1000, 500, 250)):
        st = target.tables[f"table_{i}"]
        for shard in st.local_shards():
            off = shard.metadata.shard_offsets[0]
            n = shard.tensor.shape[0]
            want = (
                torch.arange(off, off + n, dtype=torch.float32)
                .unsqueeze(1)
                .expand(n, 64)
                + i
            )
            assert torch.equal(shard.tensor, want)
    if rank == 0:
        print("restored; all table rows verified")
        import shutil

        shutil.rmtree(holder[0], ignore_errors=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
