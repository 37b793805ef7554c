#!/usr/bin/env python3
"""Sharded-embedding checkpoint benchmark — the torchrec-DLRM workload
shape (reference benchmarks/torchrec/main.py:119-235) without requiring
torchrec: row-wise ShardedTensor embedding tables (--gb-per-rank per
rank), sync take vs async take vs rank-0 torch.save, with peak RSS
reported. torchrec's DistributedModelParallel produces exactly these
ShardedTensor state dicts, so this measures the same checkpoint path.

Modes (BASELINE.json config 5: save on N ranks, reshard-restore on N/2):
    all      (default) sync + async + torch.save comparison, then cleanup
    save     sync take into --work-dir/reshard and keep it
    restore  restore --work-dir/reshard at the CURRENT world size (launch
             with a different --nproc-per-node than the save) and verify
             row content bit-exactly

Launch: python -m torch.distributed.run --nproc-per-node N \
            --master-addr 127.0.0.1 benchmarks/sharded_embedding/main.py
With --share-device every rank uses cuda:0 over gloo (1-GPU boxes).
"""

import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
)


import argparse
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


def _row_values(table_idx: int, rows: torch.Tensor) -> torch.Tensor:
    """Deterministic per-row scalar, cheap and sharding-independent."""
    return (
        ((rows.to(torch.float64) * 2654435761.0 + table_idx * 97.0) % 1000003.0)
        / 1000003.0
    ).to(torch.float32)


def fill_deterministic(st, table_idx: int) -> None:
    for shard in st.local_shards():
        off = shard.metadata.shard_offsets[0]
        n = shard.tensor.shape[0]
        rows = torch.arange(off, off + n)
        shard.tensor.copy_(
            _row_values(table_idx, rows)
            .unsqueeze(1)
            .expand(n, shard.tensor.shape[1])
            .to(shard.tensor.device)
        )


def verify_deterministic(st, table_idx: int) -> None:
    for shard in st.local_shards():
        off = shard.metadata.shard_offsets[0]
        n = shard.tensor.shape[0]
        rows = torch.arange(off, off + n)
        want = (
            _row_values(table_idx, rows)
            .unsqueeze(1)
            .expand(n, shard.tensor.shape[1])
            .to(shard.tensor.device)
        )
        assert torch.equal(shard.tensor, want), (
            f"table {table_idx} rows [{off}, {off + n}) mismatch after "
            "reshard-restore"
        )


def build_tables(
    total_gb: float,
    device: torch.device,
    num_tables: int = 4,
    deterministic: bool = False,
):
    from torch.distributed._shard import sharded_tensor
    from torch.distributed._shard.sharding_spec import ChunkShardingSpec

    world_size = dist.get_world_size()
    dev = (
        f"cuda:{device.index}" if device.type == "cuda" else "cpu"
    )
    spec = ChunkShardingSpec(
        dim=0,
        placements=[f"rank:{r}/{dev}" for r in range(world_size)],
    )
    bytes_per_table = int(total_gb * 1e9 / num_tables)
    dim = 128
    rows = bytes_per_table // (dim * 4)
    tables = {}
    for i in range(num_tables):
        st = sharded_tensor.empty(spec, (rows, dim))
        if deterministic:
            fill_deterministic(st, i)
        else:
            for shard in st.local_shards():
                shard.tensor.uniform_(-1, 1)
        tables[f"table_{i}"] = st
    total = num_tables * rows * dim * 4
    return tables, total


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gb-per-rank", type=float, default=4.0)
    parser.add_argument(
        "--total-gb",
        type=float,
        default=None,
        help="total table bytes across ranks (overrides --gb-per-rank; "
        "use for reshard runs so save and restore worlds agree)",
    )
    parser.add_argument("--num-tables", type=int, default=4)
    parser.add_argument("--work-dir", default="/tmp/tsamd_embedding_bench")
    parser.add_argument("--device", default="cuda")
    parser.add_argument(
        "--mode", choices=["all", "save", "restore"], default="all"
    )
    parser.add_argument(
        "--share-device",
        action="store_true",
        help="all ranks on cuda:0 over gloo (single-GPU boxes)",
    )
    args = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = args.device == "cuda" and torch.cuda.is_available()
    if args.share_device:
        local_rank = 0
    if use_cuda:
        torch.cuda.set_device(local_rank)
    backend = "nccl" if (use_cuda and not args.share_device) else "gloo"
    kwargs = (
        {"device_id": torch.device("cuda", local_rank)}
        if backend == "nccl"
        else {}
    )
    dist.init_process_group(backend=backend, **kwargs)
    rank = dist.get_rank()
    device = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")

    total_gb = (
        args.total_gb
        if args.total_gb is not None
        else args.gb_per_rank * world_size
    )
    deterministic = args.mode in ("save", "restore")
    tables, total_bytes = build_tables(
        total_gb, device, num_tables=args.num_tables, deterministic=deterministic
    )
    holder = _Holder(tables)

    if args.mode == "save":
        if rank == 0:
            shutil.rmtree(
                os.path.join(args.work_dir, "reshard"), ignore_errors=True
            )
            os.makedirs(args.work_dir, exist_ok=True)
        dist.barrier()
        rss = deque(maxlen=100000)
        with measure_rss_deltas(rss):
            t0 = time.monotonic()
            Snapshot.take(
                os.path.join(args.work_dir, "reshard"), {"emb": holder}
            )
            dist.barrier()
            save_s = time.monotonic() - t0
        if rank == 0:
            print(
                f"reshard-save world {world_size}: {save_s:.2f}s "
                f"({total_bytes / 1e9 / save_s:.2f} GB/s aggregate, "
                f"{total_bytes / 1e9:.1f} GB tables), "
                f"peak RSS delta {max_rss_delta_mb(rss):.0f} MB"
            )
        dist.destroy_process_group()
        return

    if args.mode == "restore":
        # zero the local shards so verification proves the restore
        for st in tables.values():
            for shard in st.local_shards():
                shard.tensor.zero_()
        rss = deque(maxlen=100000)
        with measure_rss_deltas(rss):
            t0 = time.monotonic()
            Snapshot(os.path.join(args.work_dir, "reshard")).restore(
                {"emb": holder}
            )
            dist.barrier()
            restore_s = time.monotonic() - t0
        for i in range(args.num_tables):
            verify_deterministic(holder.tables[f"table_{i}"], i)
        if rank == 0:
            print(
                f"reshard-restore world {world_size}: {restore_s:.2f}s "
                f"({total_bytes / 1e9 / restore_s:.2f} GB/s aggregate), "
                f"peak RSS delta {max_rss_delta_mb(rss):.0f} MB; "
                "row content verified bit-exact"
            )
        dist.destroy_process_group()
        return

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
