#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
export MASTER_ADDR=127.0.0.1 MASTER_PORT=29541 RANK=0 WORLD_SIZE=1 LOCAL_RANK=0
mkdir -p gpurun_out
{
  echo "=== pytest -m gpu ==="
  timeout 420 python -m pytest tests/ -q -m gpu 2>&1 | tail -3
  echo "=== async-overlap interference bench ==="
  timeout 500 python benchmarks/async_overlap/main.py --model-gb 8 2>&1 | tail -6
  rm -rf /tmp/tsamd_overlap_bench
  df -B1G / | tail -1
  echo "=== sharded_embedding 4GB 1 rank (ShardedTensor on GPU) ==="
  timeout 400 python benchmarks/sharded_embedding/main.py --gb-per-rank 4 2>&1 | tail -4
  rm -rf /tmp/tsamd_embedding_bench
  echo "=== fsdp transformer world-1 (7.8GB module on GPU) ==="
  timeout 400 python benchmarks/fsdp/main.py --benchmark-load 2>&1 | tail -3
  rm -rf /tmp/tsamd_fsdp_bench
  df -B1G / | tail -1
  echo "=== many-small-tensors stress (2000 tensors) ==="
  timeout 400 python - <<'EOF'
import torch, time, shutil
from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict
dev = torch.device("cuda", 0)
sd = StateDict(**{f"t{i}": torch.randn(512, 512, dtype=torch.bfloat16, device=dev) for i in range(2000)})
total = sum(t.numel()*t.element_size() for t in sd.values())/1e9
path = "/tmp/tsamd_many/snap"
shutil.rmtree("/tmp/tsamd_many", ignore_errors=True)
t0 = time.monotonic()
snap = Snapshot.take(path, {"sd": sd})
dt = time.monotonic() - t0
print(f"2000-tensor save ({total:.2f} GB): {dt:.2f}s = {total/dt:.2f} GB/s")
out = StateDict(**{f"t{i}": torch.zeros(512, 512, dtype=torch.bfloat16, device=dev) for i in range(2000)})
t0 = time.monotonic()
snap.restore({"sd": out})
dt = time.monotonic() - t0
print(f"2000-tensor restore: {dt:.2f}s = {total/dt:.2f} GB/s")
assert torch.equal(out["t7"], sd["t7"]) and torch.equal(out["t1999"], sd["t1999"])
shutil.rmtree("/tmp/tsamd_many", ignore_errors=True)
print("verified")
EOF
  echo "=== bench sanity ==="
  timeout 400 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  echo "=== done ==="
} > gpurun_out/check7.log 2>&1
tail -45 gpurun_out/check7.log
