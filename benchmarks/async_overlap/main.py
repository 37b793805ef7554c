#!/usr/bin/env python3
"""Never-stall async snapshots: measure how much a live training loop is
perturbed by an async_take running underneath it (BASELINE.json config 4:
"async_take during a live training step — stall time vs sync take").

A bf16 GEMM loop stands in for the training step. We measure:
  - baseline step time (no checkpointing),
  - step times while a sync Snapshot.take blocks (worst case),
  - the async_take call stall + step times while its staging/storage I/O
    drains in the background (both staging modes).
"""

import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
)


import argparse
import shutil
import time

import torch


def step_times(fn_steps: int, a, b) -> list:
    times = []
    for _ in range(fn_steps):
        torch.cuda.synchronize()
        t0 = time.monotonic()
        for _ in range(8):
            c = a @ b
        torch.cuda.synchronize()
        times.append(time.monotonic() - t0)
    return times


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--work-dir", default="/tmp/tsamd_overlap_bench")
    parser.add_argument("--model-gb", type=float, default=8.0)
    args = parser.parse_args()

    from torchsnapshot_amd import Snapshot
    from torchsnapshot_amd.state_dict import StateDict

    device = torch.device("cuda", 0)
    n = 8192
    a = torch.randn(n, n, dtype=torch.bfloat16, device=device)
    b = torch.randn(n, n, dtype=torch.bfloat16, device=device)

    # checkpoint payload
    n_tensors = int(args.model_gb * 1e9 / (64 * 1024 * 1024))
    sd = StateDict(
        **{
            f"w{i}": torch.randn(
                32, 1024, 1024, dtype=torch.bfloat16, device=device
            )
            for i in range(n_tensors)
        }
    )
    total = sum(t.numel() * t.element_size() for t in sd.values()) / 1e9
    shutil.rmtree(args.work_dir, ignore_errors=True)
    path = os.path.join(args.work_dir, "snap")

    base = step_times(6, a, b)
    base_ms = sorted(base)[len(base) // 2] * 1000
    print(f"baseline GEMM step: {base_ms:.1f} ms (payload {total:.1f} GB)")

    # sync take: training fully blocked
    t0 = time.monotonic()
    Snapshot.take(path, {"sd": sd})
    sync_s = time.monotonic() - t0
    print(f"sync take: blocks training for {sync_s:.2f}s")

    for mode in ("slab", "direct"):
        os.environ["TSAMD_STAGE_MODE"] = mode
        t0 = time.monotonic()
        pending = Snapshot.async_take(path, {"sd": sd})
        stall_s = time.monotonic() - t0
        during = step_times(10, a, b)
        pending.wait()
        after = step_times(4, a, b)
        during_ms = sorted(during)[len(during) // 2] * 1000
        after_ms = sorted(after)[len(after) // 2] * 1000
        print(
            f"async take [{mode}]: stall {stall_s:.2f}s; GEMM step during "
            f"drain {during_ms:.1f} ms ({during_ms / base_ms:.2f}x baseline), "
            f"after {after_ms:.1f} ms"
        )

    shutil.rmtree(args.work_dir, ignore_errors=True)


if __name__ == "__main__":
    main()
