#!/bin/bash
set -x
export PYTHONPATH=$GRAFT_REPO_ROOT
mkdir -p gpurun_out
{
  echo "=== pytest -m gpu with durations ==="
  timeout 420 python -m pytest tests/ -q -m gpu --durations=6 > gpurun_out/pytest_gpu10.txt 2>&1
  echo "PYTEST_RC=$?"
  grep -E "passed|failed|durations|s call" gpurun_out/pytest_gpu10.txt | head -10
  echo "=== smoke() ==="
  timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -2
  echo "=== bench final (grid-256 direct default) ==="
  timeout 500 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
  rm -rf /tmp/tsamd_bench
  echo "=== rocprof csv of bench (1 step) ==="
  cd /tmp && export TMPDIR=/tmp
  TSAMD_BENCH_DIR=/tmp/tsamd_profbench timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d $GRAFT_REPO_ROOT/gpurun_out/prof10 -o bench -- python $GRAFT_REPO_ROOT/bench.py --gpus 1 --steps 1 --warmup 1 2>&1 | tail -2
  rm -rf /tmp/tsamd_profbench
  cd $GRAFT_REPO_ROOT
  find gpurun_out/prof10 -type f
  for f in $(find gpurun_out/prof10 -name "*kernel_stats*"); do echo "--- $f"; head -12 "$f"; done
  echo "=== done ==="
} > gpurun_out/check10.log 2>&1
tail -45 gpurun_out/check10.log
