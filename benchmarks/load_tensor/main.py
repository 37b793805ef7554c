#!/usr/bin/env python3
"""Single-tensor load benchmark (reference benchmarks/load_tensor/
main.py:26-63): save a 10 GB f32 tensor, then load it back into the GPU
with and without a 100 MB memory budget, reporting wall time and peak
RSS delta — demonstrating that tiled byte-range reads bound host memory.
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

from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.rss_profiler import measure_rss_deltas, max_rss_delta_mb
from torchsnapshot_amd.state_dict import StateDict


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--dim", type=int, default=50000)  # 10 GB f32
    parser.add_argument("--memory-budget-mb", type=int, default=100)
    parser.add_argument("--work-dir", default="/tmp/tsamd_load_tensor")
    parser.add_argument("--device", default="cuda")
    args = parser.parse_args()

    device = (
        torch.device("cuda", 0)
        if args.device == "cuda" and torch.cuda.is_available()
        else torch.device("cpu")
    )
    shutil.rmtree(args.work_dir, ignore_errors=True)
    path = os.path.join(args.work_dir, "snapshot")

    t = torch.empty(args.dim, args.dim, dtype=torch.float32, device=device)
    t.uniform_(-1, 1)
    nbytes = t.numel() * 4
    Snapshot.take(path, {"sd": StateDict(t=t)})
    print(f"saved {nbytes / 1e9:.1f} GB")

    for budget in (None, args.memory_budget_mb * 1024 * 1024):
        out = torch.empty_like(t)
        rss = deque(maxlen=10000)
        with measure_rss_deltas(rss):
            t0 = time.monotonic()
            snap = Snapshot(path)
            loaded = snap.read_object(
                "0/sd/t", obj_out=out, memory_budget_bytes=budget
            )
            elapsed = time.monotonic() - t0
        assert loaded is out
        label = f"{budget // 1024 // 1024} MB budget" if budget else "no budget"
        print(
            f"load ({label}): {elapsed:.2f}s "
            f"({nbytes / 1e9 / elapsed:.2f} GB/s), peak RSS delta "
            f"{max_rss_delta_mb(rss):.0f} MB"
        )
        assert torch.equal(out, t)

    shutil.rmtree(args.work_dir, ignore_errors=True)


if __name__ == "__main__":
    main()
