#!/bin/bash
# Round-2 cold-path I/O measurement on the MI355X box:
#  1. raw NVMe read rate (cold) as the ceiling
#  2. cold restore: single-stream vs parallel-segment reads (sweep)
#  3. durable (fsync) save: single-stream vs parallel-segment writes
# Writes results to gpurun_out/r02_cold_io.log
set -u
LOG=gpurun_out/r02_cold_io.log
mkdir -p gpurun_out
: > "$LOG"
BD=/tmp/tsamd_coldio
rm -rf "$BD"; mkdir -p "$BD"

echo "== disk ==" >> "$LOG"
df -h /tmp >> "$LOG" 2>&1

# build a 16 GB snapshot once (the llama bench state, 1 warmup 0 steps is
# not possible; use steps 1 warmup 0 and keep)
python bench.py --gpus 1 --steps 1 --warmup 0 --dir "$BD" --keep >> "$LOG" 2>&1

drop_caches() { sync; echo 3 > /proc/sys/vm/drop_caches; }

echo "== raw cold read rate (dd 64M blocks over payload files) ==" >> "$LOG"
drop_caches
python - "$BD/ckpt" >> "$LOG" 2>&1 <<'EOF'
import os, sys, time
root = sys.argv[1]
files = []
for d, _, fs in os.walk(root):
    files += [os.path.join(d, f) for f in fs]
total = sum(os.path.getsize(f) for f in files)
t0 = time.monotonic()
buf = bytearray(64 * 1024 * 1024)
for f in files:
    fd = os.open(f, os.O_RDONLY)
    while os.readv(fd, [buf]) > 0:
        pass
    os.close(fd)
dt = time.monotonic() - t0
print(f"raw read: {total/1e9:.2f} GB in {dt:.2f}s = {total/1e9/dt:.2f} GB/s")
EOF

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
print(f"cold restore: {total/1e9:.2f} GB in {dt:.2f}s = {total/1e9/dt:.2f} GB/s  "
      f"(seg={os.environ.get('TSAMD_FS_IO_SEGMENT_BYTES','default')}, "
      f"min={os.environ.get('TSAMD_FS_PARALLEL_IO_MIN_BYTES','default')})")
EOF
}

echo "== cold restore: single-stream (parallel disabled) ==" >> "$LOG"
TSAMD_FS_PARALLEL_IO_MIN_BYTES=999999999999 restore_timed
echo "== cold restore: parallel segments 32M ==" >> "$LOG"
TSAMD_FS_IO_SEGMENT_BYTES=33554432 restore_timed
echo "== cold restore: parallel segments 64M (default) ==" >> "$LOG"
restore_timed
echo "== cold restore: parallel segments 128M ==" >> "$LOG"
TSAMD_FS_IO_SEGMENT_BYTES=134217728 restore_timed

save_timed() {
  rm -rf "$BD/fsync"
  python - "$BD/fsync" >> "$LOG" 2>&1 <<'EOF'
import os, sys, time, torch
sys.path.insert(0, os.getcwd())
from bench import build_state
from torchsnapshot_amd import Snapshot
state, total = build_state(torch.device("cuda", 0), 1, torch.bfloat16)
torch.cuda.synchronize()
t0 = time.monotonic()
Snapshot.take(sys.argv[1], {"model": state})
torch.cuda.synchronize()
dt = time.monotonic() - t0
print(f"durable save: {total/1e9:.2f} GB in {dt:.2f}s = {total/1e9/dt:.2f} GB/s  "
      f"(seg={os.environ.get('TSAMD_FS_IO_SEGMENT_BYTES','default')}, "
      f"min={os.environ.get('TSAMD_FS_PARALLEL_IO_MIN_BYTES','default')})")
EOF
}

echo "== durable save (TSAMD_FSYNC=1): single-stream ==" >> "$LOG"
TSAMD_FSYNC=1 TSAMD_FS_PARALLEL_IO_MIN_BYTES=999999999999 save_timed
echo "== durable save: parallel segments 64M ==" >> "$LOG"
TSAMD_FSYNC=1 save_timed
echo "== durable save: parallel segments 128M ==" >> "$LOG"
TSAMD_FSYNC=1 TSAMD_FS_IO_SEGMENT_BYTES=134217728 save_timed

rm -rf "$BD"
echo done >> "$LOG"
tail -40 "$LOG"
