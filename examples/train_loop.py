#!/usr/bin/env python3
"""Canonical production pattern: periodic non-blocking checkpoints inside
a training loop. The previous pending snapshot is awaited before the next
one starts; training only ever stalls for the staging time (sub-second
for multi-GB models on MI355X)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import tempfile
import time

import torch

from torchsnapshot_amd import PendingSnapshot, RNGState, Snapshot, StateDict

device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
model = torch.nn.Sequential(
    torch.nn.Linear(512, 2048), torch.nn.GELU(), torch.nn.Linear(2048, 512)
).to(device)
optim = torch.optim.AdamW(model.parameters(), lr=1e-3)
progress = StateDict(step=0)
app_state = {"model": model, "optim": optim, "progress": progress, "rng": RNGState()}

ckpt_root = tempfile.mkdtemp()
pending: PendingSnapshot | None = None

for step in range(1, 31):
    x = torch.randn(32, 512, device=device)
    loss = (model(x) ** 2).mean()
    loss.backward()
    optim.step()
    optim.zero_grad()
    progress["step"] = step

    if step % 10 == 0:
        if pending is not None:
            pending.wait()  # a no-op if storage I/O already drained
        t0 = time.monotonic()
        pending = Snapshot.async_take(f"{ckpt_root}/step_{step}", app_state)
        print(
            f"step {step}: checkpoint staged in "
            f"{time.monotonic() - t0:.3f}s (training continues)"
        )

if pending is not None:
    snapshot = pending.wait()
    print("last snapshot:", snapshot.path)
    # resume check
    progress["step"] = -1
    snapshot.restore(app_state)
    print("restored step:", progress["step"])
