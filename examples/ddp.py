#!/usr/bin/env python3
"""DDP example: replicated state is detected automatically, written once,
and the write load spreads across ranks.

Launch: python -m torch.distributed.run --nproc-per-node 2 \
            --master-addr 127.0.0.1 examples/ddp.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import os
import tempfile

import torch
import torch.distributed as dist
from torch.nn.parallel import DistributedDataParallel as DDP

from torchsnapshot_amd import Snapshot, StateDict


def main() -> None:
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
        device = torch.device("cuda", torch.cuda.current_device())
    else:
        device = torch.device("cpu")

    model = DDP(torch.nn.Linear(1024, 1024).to(device))
    optim = torch.optim.SGD(model.parameters(), lr=0.1)
    progress = StateDict(step=0)

    model(torch.rand(8, 1024, device=device)).sum().backward()
    optim.step()
    progress["step"] += 1

    tmp = [tempfile.mkdtemp() if rank == 0 else None]
    dist.broadcast_object_list(tmp, src=0)
    path = os.path.join(tmp[0], "snap")

    # async: training resumes once device state is staged
    pending = Snapshot.async_take(path, {"model": model, "optim": optim, "progress": progress})
    # ... next training step could run here ...
    snapshot = pending.wait()
    if rank == 0:
        print("saved:", snapshot.path)

    snapshot.restore({"model": model, "optim": optim, "progress": progress})
    if rank == 0:
        print("restored; step =", progress["step"])
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
