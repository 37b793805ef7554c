#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== 2000-tensor restore (pinned spans) ==="
  timeout 400 python - <<'PYEOF'
import torch, time, shutil
from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict
dev = torch.device("cuda", 0)
sd = StateDict(**{f"t{i}": torch.randn(512, 512, dtype=torch.bfloat16, device=dev) for i in range(2000)})
total = sum(t.numel()*t.element_size() for t in sd.values())/1e9
path = "/tmp/tsamd_many/snap"
shutil.rmtree("/tmp/tsamd_many", ignore_errors=True)
snap = Snapshot.take(path, {"sd": sd})
out = StateDict(**{f"t{i}": torch.zeros(512, 512, dtype=torch.bfloat16, device=dev) for i in range(2000)})
for trial in range(3):
    t0 = time.monotonic(); snap.restore({"sd": out}); dt = time.monotonic()-t0
    print(f"restore: {dt:.2f}s = {total/dt:.2f} GB/s")
assert torch.equal(out["t0"], sd["t0"]) and torch.equal(out["t1999"], sd["t1999"])
shutil.rmtree("/tmp/tsamd_many", ignore_errors=True)
print("verified")
PYEOF
  echo "=== checksummed bench (xxh3 overhead on 16GB) ==="
  TSAMD_CHECKSUM=1 timeout 500 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  echo "=== checksummed restore verify (16GB) ==="
  TSAMD_CHECKSUM=1 TSAMD_VERIFY_CHECKSUM=1 timeout 500 python - <<'PYEOF'
import torch, time, shutil
from torchsnapshot_amd import Snapshot
from bench import build_state
dev = torch.device("cuda", 0)
state, total = build_state(dev, 1, torch.bfloat16)
path = "/tmp/tsamd_ck/snap"
shutil.rmtree("/tmp/tsamd_ck", ignore_errors=True)
t0 = time.monotonic(); Snapshot.take(path, {"model": state}); t1 = time.monotonic()
print(f"checksummed take: {total/1e9/(t1-t0):.2f} GB/s")
state2, _ = build_state(dev, 1, torch.bfloat16)
t0 = time.monotonic(); Snapshot(path).restore({"model": state2}); dt = time.monotonic()-t0
print(f"verified restore: {total/1e9/dt:.2f} GB/s")
shutil.rmtree("/tmp/tsamd_ck", ignore_errors=True)
PYEOF
  rm -rf /tmp/tsamd_ck
  echo "=== done ==="
} > gpurun_out/check15.log 2>&1
tail -20 gpurun_out/check15.log
