#!/usr/bin/env python3
"""FSDP checkpoint benchmark (reference benchmarks/fsdp/main.py:36-135):
a ~1.9B-param transformer sharded with FSDP2 (fully_shard -> DTensor
state dicts), save + optional load, vs rank-0 torch.save of the full
state.

Launch: python -m torch.distributed.run --nproc-per-node N \
            --master-addr 127.0.0.1 benchmarks/fsdp/main.py
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


def build_transformer(device: torch.device) -> torch.nn.Module:
    # ~1.9B params like the reference's nn.Transformer config
    model = torch.nn.Transformer(
        d_model=2048,
        nhead=16,
        num_encoder_layers=12,
        num_decoder_layers=12,
        dim_feedforward=8192,
        device=device,
    )
    return model


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--work-dir", default="/tmp/tsamd_fsdp_bench")
    parser.add_argument("--benchmark-load", action="store_true")
    parser.add_argument("--compare-torch-save", action="store_true")
    args = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    torch.cuda.set_device(local_rank)
    if world_size > 1:
        dist.init_process_group(backend="nccl")
    device = torch.device("cuda", local_rank)

    model = build_transformer(device)
    if world_size > 1:
        from torch.distributed.fsdp import fully_shard

        for layer in list(model.encoder.layers) + list(model.decoder.layers):
            fully_shard(layer)
        fully_shard(model)

    nbytes = sum(p.numel() * p.element_size() for p in model.parameters())
    path = os.path.join(args.work_dir, "snapshot")
    if rank == 0:
        shutil.rmtree(args.work_dir, ignore_errors=True)
        os.makedirs(args.work_dir, exist_ok=True)
    if world_size > 1:
        dist.barrier()

    t0 = time.monotonic()
    snapshot = Snapshot.take(path, {"model": model})
    if world_size > 1:
        dist.barrier()
    elapsed = time.monotonic() - t0
    if rank == 0:
        print(
            f"take: {elapsed:.2f}s ({nbytes / 1e9 / elapsed:.2f} GB/s model "
            f"of {nbytes / 1e9:.1f} GB, {world_size} ranks)"
        )

    if args.benchmark_load:
        t0 = time.monotonic()
        snapshot.restore({"model": model})
        if world_size > 1:
            dist.barrier()
        elapsed = time.monotonic() - t0
        if rank == 0:
            print(f"restore: {elapsed:.2f}s ({nbytes / 1e9 / elapsed:.2f} GB/s)")

    if args.compare_torch_save and world_size == 1 and rank == 0:
        t0 = time.monotonic()
        torch.save(
            model.state_dict(), os.path.join(args.work_dir, "torch_save.pt")
        )
        print(f"torch.save: {time.monotonic() - t0:.2f}s")

    if rank == 0:
        shutil.rmtree(args.work_dir, ignore_errors=True)
    if world_size > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
