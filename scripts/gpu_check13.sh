#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== memcpy+kernel timeline of one bench step ==="
  cd /tmp && export TMPDIR=/tmp
  TSAMD_BENCH_DIR=/tmp/tsamd_tl timeout 500 rocprofv3 --kernel-trace --memory-copy-trace --stats --output-format csv -d $GRAFT_REPO_ROOT/gpurun_out/tl -o bench -- python $GRAFT_REPO_ROOT/bench.py --gpus 1 --steps 1 --warmup 1 2>&1 | grep -E "metric" | tail -1
  rm -rf /tmp/tsamd_tl
  cd $GRAFT_REPO_ROOT
  find gpurun_out/tl -type f
  echo "--- memory copy stats"
  for f in $(find gpurun_out/tl -name "*memory_copy_stats*"); do cat "$f"; done
  echo "=== slab vs direct one more time (fresh box state) ==="
  TSAMD_STAGE_MODE=slab timeout 400 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  TSAMD_STAGE_MODE=direct timeout 400 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  echo "=== done ==="
} > gpurun_out/check13.log 2>&1
tail -25 gpurun_out/check13.log
