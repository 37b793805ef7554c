#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== pytest -m gpu (exit code explicit) ==="
  timeout 420 python -m pytest tests/ -q -m gpu > gpurun_out/pytest_gpu.txt 2>&1
  echo "PYTEST_RC=$?"
  grep -E "passed|failed" gpurun_out/pytest_gpu.txt | tail -2
  echo "=== direct-mode grid sweep: GB/s at small grids ==="
  timeout 400 python - <<'PYEOF'
import torch, time, os
from torchsnapshot_amd.ops import staging
os.environ["TSAMD_STAGE_MODE"] = "direct"
dev = torch.device("cuda", 0)
eng = staging.get_staging_engine(dev)
payload = [torch.empty(128*1024*1024, device=dev).normal_() for _ in range(4)]  # 2GB contig x4... non-single so pack path
for grid in (16384, 1024, 256, 128, 64, 32):
    os.environ["TSAMD_PACK_GRID"] = str(grid)
    for trial in range(2):
        b = eng.stage(payload); b.wait(); b.release()
    t0 = time.monotonic()
    b = eng.stage(payload); b.wait()
    dt = time.monotonic() - t0
    b.release()
    print(f"grid {grid:6d}: {2.0/dt:.2f} GB/s")
PYEOF
  echo "=== interference at grid 128 vs 16384 (GEMM during async drain) ==="
  timeout 500 python - <<'PYEOF'
import torch, time, os, shutil
from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict
os.environ["TSAMD_STAGE_MODE"] = "direct"
dev = torch.device("cuda", 0)
n = 8192
A = torch.randn(n, n, dtype=torch.bfloat16, device=dev)
B = torch.randn(n, n, dtype=torch.bfloat16, device=dev)
sd = StateDict(**{f"w{i}": torch.randn(32, 1024, 1024, dtype=torch.bfloat16, device=dev) for i in range(128)})  # 8GB
def gemm_ms(k=10):
    ts = []
    for _ in range(k):
        torch.cuda.synchronize(); t0 = time.monotonic()
        for _ in range(8): C = A @ B
        torch.cuda.synchronize(); ts.append(time.monotonic()-t0)
    return sorted(ts)[len(ts)//2]*1000
base = gemm_ms(6)
print(f"baseline {base:.1f} ms")
for grid in ("16384", "128"):
    os.environ["TSAMD_PACK_GRID"] = grid
    shutil.rmtree("/tmp/tsamd_ov", ignore_errors=True)
    t0 = time.monotonic()
    pending = Snapshot.async_take("/tmp/tsamd_ov/snap", {"sd": sd})
    stall = time.monotonic()-t0
    during = gemm_ms(10)
    pending.wait()
    print(f"grid {grid}: stall {stall:.2f}s, during {during:.1f} ms ({during/base:.2f}x)")
shutil.rmtree("/tmp/tsamd_ov", ignore_errors=True)
PYEOF
  echo "=== PMC counters on gather kernel ==="
  cd /tmp && export TMPDIR=/tmp
  timeout 400 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/pmc -o pack --output-format csv -- python - <<'PYEOF' 2>&1 | tail -3
import torch, os
os.environ["TSAMD_STAGE_MODE"] = "slab"
from torchsnapshot_amd.ops import staging
dev = torch.device("cuda", 0)
eng = staging.get_staging_engine(dev)
wide = [torch.empty(2048, 4096, device=dev).normal_().t() for _ in range(8)]
contig = [torch.empty(64*1024*1024, device=dev).normal_() for _ in range(4)]
for group in (wide, contig):
    b = eng.stage(group); b.wait(); b.release()
print("done")
PYEOF
  cd $GRAFT_REPO_ROOT
  find gpurun_out/pmc -type f | head
  for f in $(find gpurun_out/pmc -name "*counter*" -o -name "*stats*" | head -3); do echo "--- $f"; head -8 "$f"; done
  echo "=== done ==="
} > gpurun_out/check9.log 2>&1
tail -55 gpurun_out/check9.log
