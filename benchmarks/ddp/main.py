#!/usr/bin/env python3
"""DDP checkpoint benchmark (the reference's headline table workload,
benchmarks/ddp/main.py there): a fully-replicated model of
--num-params tensors x --param-size f32 elements, saved with
replicated=["**"] (write load spread over all ranks) vs a rank-0
torch.save of the same state.

Launch: python -m torch.distributed.run --nproc-per-node N \
            --master-addr 127.0.0.1 benchmarks/ddp/main.py
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


def main() -> None:
    parser = argparse.ArgumentParser()
    # param-size is in BYTES, matching the reference exactly (its model
    # builds torch.rand(param_size / 4) f32 elements per parameter,
    # reference benchmarks/ddp/main.py:25,38): 100 MB x 200 = 20 GB
    parser.add_argument("--param-size", type=int, default=int(1e8))
    parser.add_argument("--num-params", type=int, default=200)
    parser.add_argument("--work-dir", default="/tmp/tsamd_ddp_bench")
    parser.add_argument("--device", default="cuda")
    parser.add_argument("--compare-torch-save", action="store_true")
    parser.add_argument("--benchmark-load", action="store_true")
    args = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = args.device == "cuda" and torch.cuda.is_available()
    if world_size > 1:
        if use_cuda:
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend="nccl" if use_cuda else "gloo")
    device = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")

    # identical content on every rank = a DDP-trained model's state
    torch.manual_seed(42)
    sd = StateDict()
    total_bytes = 0
    for i in range(args.num_params):
        t = torch.empty(
            args.param_size // 4, dtype=torch.float32, device=device
        )
        t.uniform_(-1, 1)
        sd[f"param_{i}"] = t
        total_bytes += t.numel() * 4

    path = os.path.join(args.work_dir, "snapshot")
    if rank == 0:
        shutil.rmtree(args.work_dir, ignore_errors=True)
        os.makedirs(args.work_dir, exist_ok=True)
    if world_size > 1:
        dist.barrier()

    t0 = time.monotonic()
    Snapshot.take(path, {"model": sd}, replicated=["**"])
    if world_size > 1:
        dist.barrier()
    elapsed = time.monotonic() - t0
    if rank == 0:
        print(
            f"torchsnapshot_amd take: {elapsed:.2f}s "
            f"({total_bytes / 1e9 / elapsed:.2f} GB/s aggregate, "
            f"{world_size} ranks)"
        )

    if args.benchmark_load:
        t0 = time.monotonic()
        Snapshot(path).restore({"model": sd})
        if world_size > 1:
            dist.barrier()
        elapsed = time.monotonic() - t0
        if rank == 0:
            print(
                f"torchsnapshot_amd restore: {elapsed:.2f}s "
                f"({total_bytes / 1e9 / elapsed:.2f} GB/s aggregate)"
            )

    if args.compare_torch_save and rank == 0:
        # free the snapshot's disk space first: the torch.save copy of a
        # 20 GB model must not race the snapshot for a small local disk
        shutil.rmtree(path, ignore_errors=True)
        t0 = time.monotonic()
        torch.save(sd.state_dict(), os.path.join(args.work_dir, "torch_save.pt"))
        elapsed = time.monotonic() - t0
        print(
            f"torch.save (rank 0): {elapsed:.2f}s "
            f"({total_bytes / 1e9 / elapsed:.2f} GB/s)"
        )

    if rank == 0:
        shutil.rmtree(args.work_dir, ignore_errors=True)
    if world_size > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
