#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== first-checkpoint stall with background pool warm ==="
  timeout 500 python - <<'PYEOF'
import torch, time, shutil
from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict

dev = torch.device("cuda", 0)
# engine creation (and the warm thread) starts with the first device op
sd = StateDict(**{f"w{i}": torch.randn(32, 1024, 1024, dtype=torch.bfloat16, device=dev) for i in range(128)})  # 8GB
from torchsnapshot_amd.ops.staging import get_staging_engine
get_staging_engine(dev)  # kick the warm thread
time.sleep(5)  # model setup time in a real job
t0 = time.monotonic()
pending = Snapshot.async_take("/tmp/tsamd_first/snap", {"sd": sd})
stall = time.monotonic() - t0
pending.wait()
print(f"FIRST async_take (8GB, pool pre-warmed in background): stall {stall:.2f}s")
shutil.rmtree("/tmp/tsamd_first", ignore_errors=True)
PYEOF
  rm -rf /tmp/tsamd_first
  echo "=== delete on GPU-written snapshot ==="
  timeout 300 python - <<'PYEOF'
import torch, os
from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict
sd = StateDict(w=torch.randn(512, 512, device="cuda"))
snap = Snapshot.take("/tmp/tsamd_del/snap", {"sd": sd})
assert os.path.exists("/tmp/tsamd_del/snap/.snapshot_metadata")
snap.delete()
assert not os.path.exists("/tmp/tsamd_del/snap")
print("delete OK")
PYEOF
  echo "=== bench x2 (fresh defaults) ==="
  for i in 1 2; do
    timeout 500 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
    rm -rf /tmp/tsamd_bench
  done
  echo "=== done ==="
} > gpurun_out/check12.log 2>&1
tail -20 gpurun_out/check12.log
