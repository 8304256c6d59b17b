#!/bin/bash
set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv -d "$R/gpurun_out/prof" -- \
  python "$R/bench.py" --steps 4 --warmup 2 --batch-gpu 32 \
  > "$R/gpurun_out/prof_bench.log" 2>&1
echo "rocprof exit=$?" >> "$R/gpurun_out/prof_bench.log"
find "$R/gpurun_out/prof" -type f -size +8M -delete
du -sh "$R/gpurun_out"
find "$R/gpurun_out/prof" -type f | head -20
