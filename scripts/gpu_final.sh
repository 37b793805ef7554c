#!/bin/bash
# Final round-1 verification: everything the driver will run, plus the
# headline benches, with explicit exit codes.
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== full pytest -m gpu ==="
  timeout 420 python -m pytest tests/ -q -m gpu > gpurun_out/pytest_gpu_final.txt 2>&1
  echo "PYTEST_RC=$?"
  grep -E "passed|failed" gpurun_out/pytest_gpu_final.txt | tail -1
  echo "=== smoke ==="
  timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1
  echo "SMOKE_RC=$?"
  echo "=== bench x2 ==="
  for i in 1 2; do
    timeout 500 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
    rm -rf /tmp/tsamd_bench
  done
  echo "=== DDP 20GB headline (fresh box first run) ==="
  timeout 400 python benchmarks/ddp/main.py 2>&1 | tail -1
  rm -rf /tmp/tsamd_ddp_bench
  echo "=== done ==="
} > gpurun_out/final.log 2>&1
tail -20 gpurun_out/final.log
