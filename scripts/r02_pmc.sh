#!/bin/bash
# NOTE (r02): PMC counter collection WEDGED on this workload (rocprofv3
# --pmc + the staging engine's hipMallocAsync/multi-stream pattern); the
# run was killed at the gpurun limit. Kernel evidence ships via
# --kernel-trace stats instead (profiles/r02_bench_kernel_stats.txt).
# Kept for reference; do not re-run without an inner timeout.
# PMC counters for the two hand-written kernels (pack gather + psum64
# verify): FETCH_SIZE/WRITE_SIZE give actual memory traffic so we can
# show the gather reads each byte once (no waste) and the psum kernel is
# HBM-read-bound as designed.
set -u
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
LOG=gpurun_out/r02_pmc.log
mkdir -p gpurun_out
: > "$LOG"

cat > /tmp/pmc_driver.py <<'EOF'
import os, sys
sys.path.insert(0, os.environ["GRAFT_REPO_ROOT"])
import torch
from torchsnapshot_amd.ops import staging

torch.manual_seed(0)
# 2 GB of mixed tensors through the gather (direct mode default)
tensors = [torch.randn(64 * 1024 * 1024 // 4, device="cuda") for _ in range(32)]
engine = staging.get_staging_engine(tensors[0].device)
batch = engine.stage(tensors, compute_checksums=True)
batch.wait()
batch.release()
# 1 GB device psum64
t = torch.randint(0, 256, (1024 * 1024 * 1024,), dtype=torch.uint8, device="cuda")
v = staging.device_psum64(t, 0)
print("done", hex(v))
EOF

rocprofv3 --pmc FETCH_SIZE WRITE_SIZE --kernel-trace -d gpurun_out/r02_pmcprof -o pmc -- \
  python /tmp/pmc_driver.py >> "$LOG" 2>&1 || echo "pmc rc=$?" >> "$LOG"
find gpurun_out/r02_pmcprof -type f >> "$LOG" 2>&1
python - >> "$LOG" 2>&1 <<'EOF'
import glob, sqlite3
dbs = glob.glob("gpurun_out/r02_pmcprof/**/*.db", recursive=True)
print("dbs:", dbs)
for db in dbs:
    con = sqlite3.connect(db)
    tables = [r[0] for r in con.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    pmc_tables = [t for t in tables if "pmc" in t.lower()]
    print(db, "pmc tables:", pmc_tables)
    for t in pmc_tables:
        try:
            rows = con.execute(f"SELECT * FROM {t} LIMIT 5").fetchall()
            cols = [d[1] for d in con.execute(f"PRAGMA table_info({t})")]
            print(t, cols, rows[:3])
        except Exception as e:
            print("err", e)
EOF
tail -30 "$LOG"
