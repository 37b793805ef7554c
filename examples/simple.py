#!/usr/bin/env python3
"""Minimal example: checkpoint a model + optimizer + progress, resume."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import tempfile

import torch

from torchsnapshot_amd import RNGState, Snapshot, StateDict

model = torch.nn.Sequential(
    torch.nn.Linear(128, 256), torch.nn.ReLU(), torch.nn.Linear(256, 10)
)
optim = torch.optim.AdamW(model.parameters(), lr=1e-3)
progress = StateDict(step=0)

app_state = {
    "model": model,
    "optim": optim,
    "progress": progress,
    "rng": RNGState(),
}

# train a few steps
for _ in range(3):
    loss = model(torch.rand(16, 128)).sum()
    loss.backward()
    optim.step()
    optim.zero_grad()
    progress["step"] += 1

with tempfile.TemporaryDirectory() as d:
    snapshot = Snapshot.take(f"{d}/step_{progress['step']}", app_state)
    print("saved:", snapshot.path)

    # simulate a restart
    progress["step"] = 0
    snapshot.restore(app_state)
    print("resumed at step", progress["step"])

    # random access without restoring everything
    w = snapshot.read_object("0/model/0.weight")
    print("first layer weight:", tuple(w.shape))
