#!/bin/bash
# Round-1 first GPU check: environment probe, gpu tests, staging
# microbench, short flagship bench.
set -x
mkdir -p gpurun_out
{
  echo "=== probe ==="
  rocm-smi --showproductname 2>/dev/null | head -5
  df -h /tmp / "$GRAFT_REPO_ROOT" 2>/dev/null
  mount | grep -E ' /tmp | / ' | head -5
  python -c "import torch; print('torch cuda:', torch.cuda.is_available(), torch.cuda.get_device_name(0))"
  echo "=== pytest -m gpu ==="
  timeout 600 python -m pytest tests/ -q -m gpu -x 2>&1 | tail -20
  echo "=== staging microbench ==="
  timeout 300 python - <<'EOF'
import torch, time
from torchsnapshot_amd.ops import staging

torch.cuda.init()
dev = torch.device("cuda", 0)
eng = staging.get_staging_engine(dev)

# 1) single contiguous 4GB: SDMA D2H into pinned
t = torch.empty(1024*1024*1024, dtype=torch.float32, device=dev).normal_()
for trial in range(3):
    t0 = time.monotonic()
    b = eng.stage([t]); b.wait()
    dt = time.monotonic() - t0
    print(f"contig 4GB D2H: {4/dt:.2f} GB/s")
    b.release()

# 2) slab pack of 128 tensors x 16MB = 2GB
small = [torch.empty(4*1024*1024, dtype=torch.float32, device=dev).normal_() for _ in range(128)]
import os
for mode in ["slab", "direct"]:
    os.environ["TSAMD_STAGE_MODE"] = mode
    for trial in range(3):
        t0 = time.monotonic()
        b = eng.stage(small); b.wait()
        dt = time.monotonic() - t0
        print(f"pack 2GB mode={mode}: {2/dt:.2f} GB/s")
        b.release()
os.environ["TSAMD_STAGE_MODE"] = "slab"

# 3) strided pack: 64 transposed 32MB tensors
tr = [torch.empty(2048, 4096, device=dev).normal_().t() for _ in range(64)]
for trial in range(2):
    t0 = time.monotonic()
    b = eng.stage(tr); b.wait()
    dt = time.monotonic() - t0
    print(f"pack strided 2GB: {2/dt:.2f} GB/s")
    b.release()

# 4) raw torch reference: t.cpu() of the 4GB tensor
for trial in range(2):
    t0 = time.monotonic(); c = t.cpu(); dt = time.monotonic() - t0
    print(f"torch .cpu() 4GB (pageable): {4/dt:.2f} GB/s")
EOF
  echo "=== fs write microbench ==="
  timeout 120 python - <<'EOF'
import os, time
buf = bytearray(1024*1024*1024)
for d in ["/tmp", os.environ.get("GRAFT_REPO_ROOT", ".")]:
    p = os.path.join(d, "tsamd_disktest.bin")
    try:
        t0 = time.monotonic()
        with open(p, "wb", buffering=0) as f:
            f.write(buf)
            f.flush(); os.fsync(f.fileno())
        dt = time.monotonic() - t0
        print(f"{d}: 1GB write+fsync {1/dt:.2f} GB/s")
        os.remove(p)
    except Exception as e:
        print(d, "FAIL", e)
EOF
  echo "=== bench.py short ==="
  df -B1G /tmp | tail -1
  timeout 600 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -5
  echo "=== done ==="
} > gpurun_out/check1.log 2>&1
tail -60 gpurun_out/check1.log
