#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== full pytest -m gpu ==="
  timeout 420 python -m pytest tests/ -q -m gpu > gpurun_out/pytest_final2.txt 2>&1
  echo "PYTEST_RC=$?"
  grep -E "passed|failed" gpurun_out/pytest_final2.txt | tail -1
  echo "=== smoke ==="
  timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | grep smoke
  echo "SMOKE_RC=$?"
  echo "=== bench x2 ==="
  for i in 1 2; do
    timeout 500 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
    rm -rf /tmp/tsamd_bench
  done
  echo "=== endurance short (5 cycles) ==="
  timeout 500 python - <<'PYEOF'
import torch, time, shutil
from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict
dev = torch.device("cuda", 0)
sd = StateDict(
    big=torch.randn(512, 1024, 1024, dtype=torch.bfloat16, device=dev),
    **{f"m{i}": torch.randn(16, 1024, 1024, dtype=torch.bfloat16, device=dev) for i in range(16)},
)
out = StateDict(
    big=torch.zeros(512, 1024, 1024, dtype=torch.bfloat16, device=dev),
    **{f"m{i}": torch.zeros(16, 1024, 1024, dtype=torch.bfloat16, device=dev) for i in range(16)},
)
total = sum(t.numel()*t.element_size() for t in sd.values())/1e9
for c in range(5):
    t0=time.monotonic(); snap=Snapshot.take("/tmp/te/s", {"sd": sd}); t1=time.monotonic()
    snap.restore({"sd": out}); t2=time.monotonic()
    print(f"cycle {c}: take {total/(t1-t0):.1f} GB/s restore {total/(t2-t1):.1f} GB/s")
assert torch.equal(out["big"], sd["big"]) and torch.equal(out["m15"], sd["m15"])
print("endurance verified")
shutil.rmtree("/tmp/te", ignore_errors=True)
PYEOF
  echo "=== S3 gpu e2e (session cleanliness) ==="
  timeout 300 python - <<'PYEOF' 2>&1 | tail -3
import sys
sys.path.insert(0, "tests")
import torch
from test_s3_plugin import FakeS3
from torchsnapshot_amd import Snapshot, StateDict
server = FakeS3()
opts = {"endpoint_url": f"http://127.0.0.1:{server.port}",
        "access_key_id": "ak", "secret_access_key": "sk", "region": "r"}
sd = StateDict(w=torch.randn(1024, 1024, dtype=torch.bfloat16, device="cuda"))
snap = Snapshot.take("s3://bkt/g2", {"sd": sd}, storage_options=opts)
out = StateDict(w=torch.zeros(1024, 1024, dtype=torch.bfloat16, device="cuda"))
Snapshot("s3://bkt/g2", storage_options=opts).restore({"sd": out})
assert torch.equal(out["w"], sd["w"])
print("S3 GPU e2e OK (watch: no unclosed-session warnings expected)")
PYEOF
  echo "=== done ==="
} > gpurun_out/final2.log 2>&1
tail -30 gpurun_out/final2.log
