#!/usr/bin/env python3
"""ZeRO-3-style checkpoint benchmark (reference benchmarks/deepspeed_opt/
main.py:28-31,82-128, which needs deepspeed — not available offline).

ZeRO-3 partitions every parameter + optimizer state across ranks as flat
per-rank shards; to torchsnapshot(-amd) these are ordinary per-rank
entries. This benchmark reproduces that state shape for an OPT-30B-like
config (48 layers, hidden 7168): each rank holds 1/world_size of the
fp16 params and fp32 optimizer moments as flat tensors.
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

import torch
import torch.distributed as dist

from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict


def opt30b_param_count(layers: int = 48, hidden: int = 7168) -> int:
    per_layer = 4 * hidden * hidden + 2 * hidden * 4 * hidden + 9 * hidden
    embed = 50272 * hidden + 2050 * hidden
    return layers * per_layer + embed


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--layers", type=int, default=48)
    parser.add_argument("--hidden", type=int, default=7168)
    parser.add_argument("--work-dir", default="/tmp/tsamd_zero3_bench")
    parser.add_argument("--device", default="cuda")
    parser.add_argument("--benchmark-load", action="store_true")
    args = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = args.device == "cuda" and torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank)
    if world_size > 1:
        dist.init_process_group(backend="nccl" if use_cuda else "gloo")
    rank = int(os.environ.get("RANK", "0"))
    device = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")

    n_params = opt30b_param_count(args.layers, args.hidden)
    per_rank = n_params // world_size
    sd = StateDict(
        fp16_partition=torch.empty(per_rank, dtype=torch.float16, device=device).normal_(),
        exp_avg=torch.empty(per_rank, dtype=torch.float32, device=device).normal_(),
        exp_avg_sq=torch.empty(per_rank, dtype=torch.float32, device=device).normal_(),
        fp32_partition=torch.empty(per_rank, dtype=torch.float32, device=device).normal_(),
    )
    total_bytes = world_size * per_rank * (2 + 4 + 4 + 4)
    path = os.path.join(args.work_dir, "snapshot")
    if rank == 0:
        shutil.rmtree(args.work_dir, ignore_errors=True)
        os.makedirs(args.work_dir, exist_ok=True)
    if world_size > 1:
        dist.barrier()

    t0 = time.monotonic()
    snapshot = Snapshot.take(path, {"zero": sd})
    if world_size > 1:
        dist.barrier()
    elapsed = time.monotonic() - t0
    if rank == 0:
        print(
            f"ZeRO-3 state take: {elapsed:.2f}s "
            f"({total_bytes / 1e9 / elapsed:.2f} GB/s aggregate of "
            f"{total_bytes / 1e9:.1f} GB)"
        )

    if args.benchmark_load:
        t0 = time.monotonic()
        snapshot.restore({"zero": sd})
        if world_size > 1:
            dist.barrier()
        elapsed = time.monotonic() - t0
        if rank == 0:
            print(f"restore: {elapsed:.2f}s ({total_bytes / 1e9 / elapsed:.2f} GB/s)")

    if rank == 0:
        shutil.rmtree(args.work_dir, ignore_errors=True)
    if world_size > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
