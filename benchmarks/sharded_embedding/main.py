#!/usr/bin/env python3
"""Sharded-embedding checkpoint benchmark — the torchrec-DLRM workload
shape (reference benchmarks/torchrec/main.py:119-235) without requiring
torchrec: row-wise ShardedTensor embedding tables (--gb-per-rank per
rank), sync take vs async take vs rank-0 torch.save, with peak RSS
reported. torchrec's DistributedModelParallel produces exactly these
ShardedTensor state dicts, so this measures the same checkpoint path.

Launch: python -m torch.distributed.run --nproc-per-node N \
            --master-addr 127.0.0.1 benchmarks/sharded_embedding/main.py
"""

import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
)


import argparse
import os
import shutil
import time
from collections import deque

import torch
import torch.distributed as dist

from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.rss_profiler import measure_rss_deltas, max_rss_delta_mb


class _Holder:
    def __init__(self, tables):
        self.tables = tables

    def state_dict(self):
        return dict(self.tables)

    def load_state_dict(self, sd):
        self.tables = dict(sd)


def build_tables(gb_per_rank: float, device: torch.device, num_tables: int = 4):
    from torch.distributed._shard import sharded_tensor
    from torch.distributed._shard.sharding_spec import ChunkShardingSpec

    world_size = dist.get_world_size()
    dev_str = f"cuda:{device.index}" if device.type == "cuda" else "cpu"
    spec = ChunkShardingSpec(
        dim=0,
        placements=[
            f"rank:{r}/{'cuda:' + str(r) if device.type == 'cuda' else 'cpu'}"
            for r in range(world_size)
        ],
    )
    bytes_per_table = int(gb_per_rank * 1e9 * world_size / num_tables)
    dim = 128
    rows = bytes_per_table // (dim * 4)
    tables = {}
    for i in range(num_tables):
        st = sharded_tensor.empty(spec, (rows, dim))
        for shard in st.local_shards():
            shard.tensor.uniform_(-1, 1)
        tables[f"table_{i}"] = st
    total = num_tables * rows * dim * 4
    return tables, total


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gb-per-rank", type=float, default=4.0)
    parser.add_argument("--work-dir", default="/tmp/tsamd_embedding_bench")
    parser.add_argument("--device", default="cuda")
    args = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = args.device == "cuda" and torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank)
    dist.init_process_group(backend="nccl" if use_cuda else "gloo")
    rank = dist.get_rank()
    device = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")

    tables, total_bytes = build_tables(args.gb_per_rank, device)
    holder = _Holder(tables)
    if rank == 0:
        shutil.rmtree(args.work_dir, ignore_errors=True)
        os.makedirs(args.work_dir, exist_ok=True)
    dist.barrier()

    # sync take
    rss = deque(maxlen=100000)
    with measure_rss_deltas(rss):
        t0 = time.monotonic()
        Snapshot.take(os.path.join(args.work_dir, "sync"), {"emb": holder})
        dist.barrier()
        sync_s = time.monotonic() - t0
    if rank == 0:
        print(
            f"sync take: {sync_s:.2f}s ({total_bytes / 1e9 / sync_s:.2f} GB/s "
            f"aggregate), peak RSS delta {max_rss_delta_mb(rss):.0f} MB"
        )

    # async take: measure the training stall, then drain
    rss = deque(maxlen=100000)
    with measure_rss_deltas(rss):
        t0 = time.monotonic()
        pending = Snapshot.async_take(
            os.path.join(args.work_dir, "async"), {"emb": holder}
        )
        stall_s = time.monotonic() - t0
        pending.wait()
        total_s = time.monotonic() - t0
    if rank == 0:
        print(
            f"async take: stall {stall_s:.2f}s, total {total_s:.2f}s, "
            f"peak RSS delta {max_rss_delta_mb(rss):.0f} MB"
        )

    # rank-0 torch.save comparison (requires gathering; save local shards)
    t0 = time.monotonic()
    torch.save(
        {k: [s.tensor.cpu() for s in v.local_shards()] for k, v in tables.items()},
        os.path.join(args.work_dir, f"torch_save_{rank}.pt"),
    )
    dist.barrier()
    ts_s = time.monotonic() - t0
    if rank == 0:
        print(f"per-rank torch.save of local shards: {ts_s:.2f}s")

    if rank == 0:
        shutil.rmtree(args.work_dir, ignore_errors=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
