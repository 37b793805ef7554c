#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== pytest -m gpu ==="
  timeout 300 python -m pytest tests/ -q -m gpu 2>&1 | tail -3
  df -B1G / | tail -1
  echo "=== restore timing (direct pinned reads) ==="
  timeout 420 python - <<'EOF'
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
    dt = time.monotonic() - t0
    print(f"restore 16GB: {total/1e9/dt:.2f} GB/s ({dt:.2f}s)")
shutil.rmtree("/tmp/tsamd_restore_bench", ignore_errors=True)
EOF
  rm -rf /tmp/tsamd_restore_bench
  df -B1G / | tail -1
  echo "=== rocprof kernel stats (short) ==="
  cd /tmp && export TMPDIR=/tmp
  timeout 300 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof4 -o staging -- python - <<'EOF' 2>&1 | tail -4
import torch, os
os.environ["TSAMD_PINNED_POOL_BYTES"] = str(2 * 1024**3)
from torchsnapshot_amd.ops import staging
dev = torch.device("cuda", 0)
eng = staging.get_staging_engine(dev)
wide = [torch.empty(2048, 4096, device=dev).normal_().t() for _ in range(8)]
narrow = [torch.empty(4096, 64, device=dev).normal_()[:, ::2] for _ in range(8)]
contig = [torch.empty(32*1024*1024, device=dev).normal_() for _ in range(4)]
for mode in ("slab", "direct"):
    os.environ["TSAMD_STAGE_MODE"] = mode
    for group in (wide, narrow, contig):
        b = eng.stage(group); b.wait(); b.release()
print("profiled ok")
EOF
  cd $GRAFT_REPO_ROOT
  find gpurun_out/prof4 -type f
  for f in $(find gpurun_out/prof4 -name "*kernel_stats*"); do echo "--- $f"; cat "$f"; done
  df -B1G / | tail -1
  echo "=== DDP reference-config bench: 20GB f32 1 GPU + torch.save ==="
  timeout 500 python benchmarks/ddp/main.py --compare-torch-save 2>&1 | tail -3
  rm -rf /tmp/tsamd_ddp_bench
  df -B1G / | tail -1
  echo "=== load_tensor 10GB budgeted ==="
  timeout 500 python benchmarks/load_tensor/main.py 2>&1 | tail -4
  rm -rf /tmp/tsamd_load_tensor
  echo "=== done ==="
} > gpurun_out/check4.log 2>&1
tail -60 gpurun_out/check4.log
