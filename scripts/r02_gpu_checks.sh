#!/bin/bash
# Round-2 GPU validation: the new multi-rank shared-device tests, a
# world-2 DTensor bench rehearsal on one GPU, and a rocprof kernel-stats
# refresh of the save path.
set -u
LOG=gpurun_out/r02_gpu_checks.log
mkdir -p gpurun_out
: > "$LOG"

echo "== gpu suite (incl. new multirank shared-device tests) ==" >> "$LOG"
timeout 900 python -m pytest tests -m gpu -q --timeout 600 2>&1 | tail -6 >> "$LOG"

echo "== bench world-2 rehearsal: DTensor save on shared cuda:0 (gloo) ==" >> "$LOG"
TSAMD_BENCH_SHARE_DEVICE=1 timeout 600 python -m torch.distributed.run \
  --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29611 \
  bench.py --gpus 2 --steps 3 --warmup 1 --model llama3-8b 2>&1 | tail -4 >> "$LOG"

echo "== rocprofv3 kernel stats: 2-step bench ==" >> "$LOG"
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
rocprofv3 --kernel-trace --stats -d gpurun_out/r02_prof -o r02bench -- \
  python bench.py --gpus 1 --steps 2 --warmup 1 >> "$LOG" 2>&1 || echo "rocprof rc=$?" >> "$LOG"
find gpurun_out/r02_prof -name "*stats*" | head -3 >> "$LOG"
for f in $(find gpurun_out/r02_prof -name "*kernel_stats*"); do
  echo "-- $f" >> "$LOG"; head -15 "$f" >> "$LOG"
done
tail -45 "$LOG"
