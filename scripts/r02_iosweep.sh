#!/bin/bash
set -u
LOG=gpurun_out/r02_iosweep.log
mkdir -p gpurun_out
: > "$LOG"
for io in 24 32 48; do
  echo "== io_concurrency=$io ==" >> "$LOG"
  TSAMD_MAX_PER_RANK_IO_CONCURRENCY=$io timeout 400 \
    python bench.py --gpus 1 --steps 10 --warmup 3 2>/dev/null \
    | grep checkpoint_save >> "$LOG" || echo failed >> "$LOG"
done
cat "$LOG"
