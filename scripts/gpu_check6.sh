#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== write-path phase breakdown (defaults) ==="
  TSAMD_TIMING=1 timeout 400 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | grep -E "timing|metric" | tail -8
  rm -rf /tmp/tsamd_bench
  echo "=== A/B: io concurrency 32 ==="
  TSAMD_MAX_PER_RANK_IO_CONCURRENCY=32 timeout 400 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  echo "=== A/B: slab threshold 512MB ==="
  TSAMD_SLAB_SIZE_THRESHOLD_BYTES=536870912 timeout 400 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  echo "=== A/B: staging threads 8 ==="
  TSAMD_NUM_STAGING_THREADS=8 timeout 400 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  echo "=== A/B: write chunk 256MB ==="
  TSAMD_FS_WRITE_CHUNK_BYTES=268435456 timeout 400 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  echo "=== A/B: combined best guess ==="
  TSAMD_MAX_PER_RANK_IO_CONCURRENCY=32 TSAMD_NUM_STAGING_THREADS=8 timeout 400 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  echo "=== done ==="
} > gpurun_out/check6.log 2>&1
tail -40 gpurun_out/check6.log
