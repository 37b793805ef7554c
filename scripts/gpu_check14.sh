#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== DDP 20GB A/B: direct vs slab, io24 vs io32 ==="
  for mode in direct slab; do
    for io in 24 32; do
      TSAMD_STAGE_MODE=$mode TSAMD_MAX_PER_RANK_IO_CONCURRENCY=$io \
        timeout 400 python benchmarks/ddp/main.py 2>&1 | tail -1
      rm -rf /tmp/tsamd_ddp_bench
    done
  done
  echo "=== DDP 20GB: larger chunks (1GB) ==="
  TSAMD_MAX_CHUNK_SIZE_BYTES=1073741824 TSAMD_PINNED_BLOCK_SIZE_BYTES=1073741824 \
    timeout 400 python benchmarks/ddp/main.py 2>&1 | tail -1
  rm -rf /tmp/tsamd_ddp_bench
  echo "=== full gpu pytest ==="
  timeout 420 python -m pytest tests/ -q -m gpu > gpurun_out/pytest_gpu14.txt 2>&1
  echo "PYTEST_RC=$?"
  grep -E "passed|failed" gpurun_out/pytest_gpu14.txt | tail -1
  echo "=== bench x3 median ==="
  for i in 1 2 3; do
    timeout 400 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
    rm -rf /tmp/tsamd_bench
  done
  echo "=== done ==="
} > gpurun_out/check14.log 2>&1
tail -30 gpurun_out/check14.log
