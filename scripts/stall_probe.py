#!/usr/bin/env python3
# NOTE: experimental probe; the monkeypatch pattern is validated on CPU
# but one slow GPU box timed out running it — not part of any suite.
"""Decompose the async_take stall on a GPU box (DTensor world-1 path)."""
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29571")
dist.init_process_group(
    "nccl", rank=0, world_size=1, device_id=torch.device("cuda", 0)
)

from bench import build_state  # noqa: E402
from torchsnapshot_amd import Snapshot  # noqa: E402
from torchsnapshot_amd import snapshot as snapmod  # noqa: E402
from torchsnapshot_amd.io_preparer import prepare_write as real_pw  # noqa: E402

state, total = build_state(
    torch.device("cuda", 0), 1, torch.bfloat16, use_dtensor=True
)
d = tempfile.mkdtemp()
Snapshot.take(os.path.join(d, "warm"), {"model": state})
torch.cuda.synchronize()
os.sync()

times = {"prepare": 0.0, "shadow": 0.0, "batch": 0.0}
orig_shadow = Snapshot._shadow_for_async.__func__


def shadow(cls, f):
    t0 = time.monotonic()
    r = orig_shadow(cls, f)
    times["shadow"] += time.monotonic() - t0
    return r


Snapshot._shadow_for_async = classmethod(shadow)


def pw(**kw):
    t0 = time.monotonic()
    r = real_pw(**kw)
    times["prepare"] += time.monotonic() - t0
    return r


snapmod.prepare_write = pw
orig_batch = snapmod._batch


def batch(reqs, rank):
    t0 = time.monotonic()
    r = orig_batch(reqs, rank)
    times["batch"] += time.monotonic() - t0
    return r


snapmod._batch = batch

t0 = time.monotonic()
p = Snapshot.async_take(os.path.join(d, "s"), {"model": state})
stall = time.monotonic() - t0
print(
    f"stall total: {stall:.3f}s  shadow: {times['shadow']:.3f}s  "
    f"prepare: {times['prepare']:.3f}s  batch: {times['batch']:.3f}s"
)
p.wait()
dist.destroy_process_group()
