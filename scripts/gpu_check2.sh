#!/bin/bash
set -x
mkdir -p gpurun_out
{
  echo "=== pytest -m gpu (full) ==="
  timeout 600 python -m pytest tests/ -q -m gpu 2>&1 | tail -8
  echo "=== bench default (slab mode) ==="
  timeout 600 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -2
  echo "=== bench direct mode ==="
  TSAMD_STAGE_MODE=direct timeout 600 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -2
  echo "=== restore timing ==="
  timeout 600 python - <<'EOF'
import torch, time, os, shutil
from torchsnapshot_amd import Snapshot
from bench import build_state
dev = torch.device("cuda", 0)
state, total = build_state(dev, 1, torch.bfloat16)
path = "/tmp/tsamd_restore_bench/ckpt"
shutil.rmtree(path, ignore_errors=True)
Snapshot.take(path, {"model": state})
# restore into a fresh GPU state
state2, _ = build_state(dev, 1, torch.bfloat16)
snap = Snapshot(path)
for i in range(2):
    t0 = time.monotonic()
    snap.restore({"model": state2})
    dt = time.monotonic() - t0
    print(f"restore 16GB: {total/1e9/dt:.2f} GB/s ({dt:.2f}s)")
shutil.rmtree(path, ignore_errors=True)
EOF
  echo "=== rocprof kernel trace of staging ==="
  cd /tmp && export TMPDIR=/tmp
  timeout 600 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof -o staging -- python - <<'EOF' 2>&1 | tail -30
import torch, os
from torchsnapshot_amd.ops import staging
dev = torch.device("cuda", 0)
eng = staging.get_staging_engine(dev)
# strided pack workload: exercises the gather kernel (wide + narrow rows)
wide = [torch.empty(2048, 4096, device=dev).normal_().t() for _ in range(16)]
narrow = [torch.empty(4096, 64, device=dev).normal_()[:, ::2] for _ in range(16)]
contig = [torch.empty(64*1024*1024, device=dev).normal_() for _ in range(4)]
for mode in ("slab", "direct"):
    os.environ["TSAMD_STAGE_MODE"] = mode
    for group in (wide, narrow, contig):
        b = eng.stage(group); b.wait(); b.release()
print("profiled ok")
EOF
  cd $GRAFT_REPO_ROOT
  ls -la gpurun_out/prof/ 2>/dev/null | head
  find gpurun_out/prof -name "*stats*" | head -5
  for f in $(find gpurun_out/prof -name "*kernel_stats*"); do echo "--- $f"; head -20 "$f"; done
  echo "=== done ==="
} > gpurun_out/check2.log 2>&1
tail -70 gpurun_out/check2.log
