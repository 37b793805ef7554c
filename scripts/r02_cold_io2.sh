#!/bin/bash
# Round-2 cold-path sweep, part 2:
#  1. PARALLEL raw read ceiling (the honest comparison for a parallel
#     restore path; part 1 measured single-stream raw at 6.44 GB/s)
#  2. restore variants: WILLNEED, 16M segments, more io threads
#  3. 50 GB embedding-table reshard: save world 2 (shared GPU) -> restore
#     world 1, with peak RSS (BASELINE.json config 5 at the largest size
#     the 79 GB box disk allows)
#  4. 10-step headline bench (checks the new parallel-write fs path)
set -u
LOG=gpurun_out/r02_cold_io2.log
mkdir -p gpurun_out
: > "$LOG"
BD=/tmp/tsamd_coldio
rm -rf "$BD"; mkdir -p "$BD"

python bench.py --gpus 1 --steps 1 --warmup 0 --dir "$BD" --keep > /dev/null 2>&1

drop_caches() { sync; echo 3 > /proc/sys/vm/drop_caches; }

echo "== raw parallel read ceiling (N threads, 64M slices) ==" >> "$LOG"
for T in 8 16 24; do
drop_caches
T=$T python - "$BD/ckpt" >> "$LOG" 2>&1 <<'EOF'
import os, sys, time
from concurrent.futures import ThreadPoolExecutor
root = sys.argv[1]
nthreads = int(os.environ["T"])
files = []
for d, _, fs in os.walk(root):
    files += [os.path.join(d, f) for f in fs]
jobs = []
for f in files:
    size = os.path.getsize(f)
    for off in range(0, size, 64 * 1024 * 1024):
        jobs.append((f, off, min(off + 64 * 1024 * 1024, size)))
total = sum(e - s for _, s, e in jobs)
def read_seg(job):
    f, s, e = job
    fd = os.open(f, os.O_RDONLY)
    try:
        os.posix_fadvise(fd, s, e - s, os.POSIX_FADV_SEQUENTIAL)
        buf = bytearray(e - s)
        mv = memoryview(buf)
        off = 0
        while off < len(mv):
            off += os.preadv(fd, [mv[off:]], s + off)
    finally:
        os.close(fd)
t0 = time.monotonic()
with ThreadPoolExecutor(nthreads) as ex:
    list(ex.map(read_seg, jobs))
dt = time.monotonic() - t0
print(f"parallel raw read x{nthreads}: {total/1e9:.2f} GB in {dt:.2f}s = {total/1e9/dt:.2f} GB/s")
EOF
done

restore_timed() {
  drop_caches
  python - "$BD/ckpt" >> "$LOG" 2>&1 <<'EOF'
import os, sys, time, torch
sys.path.insert(0, os.getcwd())
from bench import build_state
from torchsnapshot_amd import Snapshot
state, total = build_state(torch.device("cuda", 0), 1, torch.bfloat16)
torch.cuda.synchronize()
t0 = time.monotonic()
Snapshot(sys.argv[1]).restore({"model": state})
torch.cuda.synchronize()
dt = time.monotonic() - t0
env = {k: v for k, v in os.environ.items() if k.startswith("TSAMD")}
print(f"cold restore: {total/1e9:.2f} GB in {dt:.2f}s = {total/1e9/dt:.2f} GB/s  {env}")
EOF
}

echo "== cold restore variants ==" >> "$LOG"
TSAMD_FS_WILLNEED=1 restore_timed
TSAMD_FS_IO_SEGMENT_BYTES=16777216 restore_timed
TSAMD_MAX_PER_RANK_IO_CONCURRENCY=48 restore_timed
TSAMD_FS_WILLNEED=1 TSAMD_MAX_PER_RANK_IO_CONCURRENCY=48 restore_timed
rm -rf "$BD"

echo "== 50 GB embedding reshard: save world 2 (shared GPU, gloo) ==" >> "$LOG"
EMB=/tmp/tsamd_emb
rm -rf "$EMB"
timeout 900 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29601 \
  benchmarks/sharded_embedding/main.py --mode save --total-gb 50 \
  --share-device --work-dir "$EMB" 2>&1 | grep -v Warning >> "$LOG"
echo "== reshard-restore world 1 ==" >> "$LOG"
drop_caches
timeout 900 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
  --master-addr 127.0.0.1 --master-port 29602 \
  benchmarks/sharded_embedding/main.py --mode restore --total-gb 50 \
  --share-device --work-dir "$EMB" 2>&1 | grep -v Warning >> "$LOG"
rm -rf "$EMB"

echo "== 10-step headline bench (parallel-write fs path) ==" >> "$LOG"
python bench.py --gpus 1 --steps 10 --warmup 3 >> "$LOG" 2>&1

echo done >> "$LOG"
tail -30 "$LOG"
