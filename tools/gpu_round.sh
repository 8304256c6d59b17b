#!/bin/bash
# First-GPU-contact script: run under gpurun from the repo root.
# Writes everything under gpurun_out/ so it merges back.
set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
export TMPDIR=/tmp

timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu exit=$?" >> gpurun_out/pytest_gpu.log

timeout 420 python bench.py --steps 16 --warmup 4 > gpurun_out/bench_1gpu.log 2>&1
echo "bench exit=$?" >> gpurun_out/bench_1gpu.log

cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d "$R/gpurun_out/prof" -- \
  python "$R/bench.py" --steps 5 --warmup 2 > "$R/gpurun_out/prof_bench.log" 2>&1
echo "rocprof exit=$?" >> "$R/gpurun_out/prof_bench.log"
tail -3 "$R/gpurun_out/pytest_gpu.log"
tail -2 "$R/gpurun_out/bench_1gpu.log"
