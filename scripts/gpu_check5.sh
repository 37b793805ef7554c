#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== pytest -m gpu (incl UVM) ==="
  timeout 300 python -m pytest tests/ -q -m gpu 2>&1 | tail -3
  echo "=== load_tensor tiled-budget RSS (fixed) ==="
  timeout 500 python benchmarks/load_tensor/main.py 2>&1 | tail -3
  rm -rf /tmp/tsamd_load_tensor
  echo "=== restore timing breakdown ==="
  TSAMD_TIMING=1 timeout 500 python - <<'EOF'
import torch, time, shutil
from torchsnapshot_amd import Snapshot
from bench import build_state
dev = torch.device("cuda", 0)
state, total = build_state(dev, 1, torch.bfloat16)
path = "/tmp/tsamd_restore_bench/ckpt"
shutil.rmtree(path, ignore_errors=True)
Snapshot.take(path, {"model": state})
state2, _ = build_state(dev, 1, torch.bfloat16)
snap = Snapshot(path)
for i in range(2):
    t0 = time.monotonic()
    snap.restore({"model": state2})
    print(f"restore 16GB: {total/1e9/(time.monotonic()-t0):.2f} GB/s")
# cold-cache restore: drop the page cache so reads hit NVMe
import subprocess
subprocess.run("sync; echo 3 > /proc/sys/vm/drop_caches", shell=True)
t0 = time.monotonic()
snap.restore({"model": state2})
print(f"restore 16GB COLD: {total/1e9/(time.monotonic()-t0):.2f} GB/s")
shutil.rmtree("/tmp/tsamd_restore_bench", ignore_errors=True)
EOF
  rm -rf /tmp/tsamd_restore_bench
  echo "=== raw disk read (cold) ==="
  timeout 200 python - <<'EOF'
import os, time, subprocess
p = "/tmp/disk_probe.bin"
with open(p, "wb") as f:
    f.write(os.urandom(100*1024*1024) * 20)  # 2 GB
subprocess.run("sync; echo 3 > /proc/sys/vm/drop_caches", shell=True)
buf = bytearray(64*1024*1024)
fd = os.open(p, os.O_RDONLY)
t0 = time.monotonic(); total = 0
while True:
    n = os.preadv(fd, [buf], total)
    if n == 0: break
    total += n
dt = time.monotonic() - t0
print(f"cold read: {total/1e9/dt:.2f} GB/s")
os.close(fd); os.remove(p)
EOF
  echo "=== bench sanity after changes ==="
  timeout 500 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  echo "=== done ==="
} > gpurun_out/check5.log 2>&1
tail -45 gpurun_out/check5.log
