#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== pytest -m gpu ==="
  timeout 420 python -m pytest tests/ -q -m gpu 2>&1 | tail -3
  echo "=== 2000-tensor restore (span fast path) ==="
  timeout 400 python - <<'PYEOF'
import torch, time, shutil
from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict
dev = torch.device("cuda", 0)
sd = StateDict(**{f"t{i}": torch.randn(512, 512, dtype=torch.bfloat16, device=dev) for i in range(2000)})
total = sum(t.numel()*t.element_size() for t in sd.values())/1e9
path = "/tmp/tsamd_many/snap"
shutil.rmtree("/tmp/tsamd_many", ignore_errors=True)
t0 = time.monotonic(); snap = Snapshot.take(path, {"sd": sd}); dt = time.monotonic()-t0
print(f"2000-tensor save ({total:.2f} GB): {dt:.2f}s = {total/dt:.2f} GB/s")
out = StateDict(**{f"t{i}": torch.zeros(512, 512, dtype=torch.bfloat16, device=dev) for i in range(2000)})
for trial in range(2):
    t0 = time.monotonic(); snap.restore({"sd": out}); dt = time.monotonic()-t0
    print(f"2000-tensor restore: {dt:.2f}s = {total/dt:.2f} GB/s")
for i in (0, 7, 999, 1999):
    assert torch.equal(out[f"t{i}"], sd[f"t{i}"]), i
print("verified")
shutil.rmtree("/tmp/tsamd_many", ignore_errors=True)
PYEOF
  echo "=== sharded_embedding full output (world-1 torchrun) ==="
  timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 --master-addr 127.0.0.1 --master-port 29551 benchmarks/sharded_embedding/main.py --gb-per-rank 2 2>&1 | tail -12
  rm -rf /tmp/tsamd_embedding_bench
  echo "=== done ==="
} > gpurun_out/check8.log 2>&1
tail -40 gpurun_out/check8.log
