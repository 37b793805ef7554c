#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== endurance: 10 take+restore cycles, pool/RSS stability ==="
  timeout 900 python - <<'PYEOF'
import torch, time, shutil, psutil, gc
from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict
from torchsnapshot_amd.ops import staging

dev = torch.device("cuda", 0)
sd = StateDict(
    big=torch.randn(1024, 1024, 1024, dtype=torch.bfloat16, device=dev),  # 2GB chunked
    **{f"m{i}": torch.randn(16, 1024, 1024, dtype=torch.bfloat16, device=dev) for i in range(32)},  # 32x32MB slabs
)
total = sum(t.numel()*t.element_size() for t in sd.values())/1e9
out = StateDict(
    big=torch.zeros(1024, 1024, 1024, dtype=torch.bfloat16, device=dev),
    **{f"m{i}": torch.zeros(16, 1024, 1024, dtype=torch.bfloat16, device=dev) for i in range(32)},
)
proc = psutil.Process()
path = "/tmp/tsamd_endurance/snap"
rss_log = []
for cycle in range(10):
    t0 = time.monotonic()
    snap = Snapshot.take(path, {"sd": sd})
    t1 = time.monotonic()
    snap.restore({"sd": out})
    t2 = time.monotonic()
    pool = staging.get_pinned_pool()
    rss_log.append(proc.memory_info().rss/1e9)
    print(f"cycle {cycle}: take {total/(t1-t0):.1f} GB/s, restore {total/(t2-t1):.1f} GB/s, "
          f"rss {rss_log[-1]:.1f} GB, pool free={len(pool._free)} alloc={pool._allocated}")
assert torch.equal(out["big"], sd["big"]) and torch.equal(out["m31"], sd["m31"])
# RSS must plateau (pool reuse, no leaks): last 3 cycles within 1GB of each other
assert max(rss_log[-3:]) - min(rss_log[-3:]) < 1.0, rss_log
print("endurance OK")
shutil.rmtree("/tmp/tsamd_endurance", ignore_errors=True)
PYEOF
  rm -rf /tmp/tsamd_endurance
  echo "=== failure recovery: failed take must not starve the pool ==="
  timeout 400 python - <<'PYEOF'
import torch, shutil, asyncio
from unittest import mock
from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict
from torchsnapshot_amd.storage.fs import FSStoragePlugin

class Faulty(FSStoragePlugin):
    async def write(self, write_io):
        raise RuntimeError("injected")

dev = torch.device("cuda", 0)
sd = StateDict(**{f"w{i}": torch.randn(16, 1024, 1024, dtype=torch.bfloat16, device=dev) for i in range(16)})
def fake(url, storage_options=None):
    return Faulty(url.split("://")[-1], storage_options)
for i in range(3):
    with mock.patch("torchsnapshot_amd.snapshot.url_to_storage_plugin", side_effect=fake):
        try:
            Snapshot.take(f"/tmp/tsamd_fail/snap{i}", {"sd": sd})
            raise SystemExit("expected failure")
        except RuntimeError:
            pass
# after 3 failed takes, a normal take must still work (pool not starved)
snap = Snapshot.take("/tmp/tsamd_fail/ok", {"sd": sd})
out = StateDict(**{f"w{i}": torch.zeros(16, 1024, 1024, dtype=torch.bfloat16, device=dev) for i in range(16)})
snap.restore({"sd": out})
assert torch.equal(out["w0"], sd["w0"])
print("failure recovery OK")
shutil.rmtree("/tmp/tsamd_fail", ignore_errors=True)
PYEOF
  rm -rf /tmp/tsamd_fail
  echo "=== S3 (fake local server) e2e with device tensors ==="
  timeout 400 python - <<'PYEOF'
import sys
sys.path.insert(0, "tests")
import torch
from test_s3_plugin import FakeS3
from torchsnapshot_amd import Snapshot, StateDict

server = FakeS3()
opts = {
    "endpoint_url": f"http://127.0.0.1:{server.port}",
    "access_key_id": "ak", "secret_access_key": "sk", "region": "r",
}
dev = torch.device("cuda", 0)
sd = StateDict(w=torch.randn(2048, 2048, dtype=torch.bfloat16, device=dev),
               small=torch.randn(64, device=dev))
snap = Snapshot.take("s3://bkt/gpu", {"sd": sd}, storage_options=opts)
out = StateDict(w=torch.zeros(2048, 2048, dtype=torch.bfloat16, device=dev),
                small=torch.zeros(64, device=dev))
Snapshot("s3://bkt/gpu", storage_options=opts).restore({"sd": out})
assert torch.equal(out["w"], sd["w"]) and torch.equal(out["small"], sd["small"])
print("S3 GPU e2e OK")
PYEOF
  echo "=== done ==="
} > gpurun_out/check11.log 2>&1
tail -35 gpurun_out/check11.log
