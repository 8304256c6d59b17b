#!/bin/bash
set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
export TMPDIR=/tmp
cd /tmp
timeout 400 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE --kernel-trace --stats --output-format csv -d "$R/gpurun_out/pmc" -- \
  python "$R/tools/kbench.py" wgrad1 > "$R/gpurun_out/pmc_kbench.log" 2>&1
echo "exit=$?" >> "$R/gpurun_out/pmc_kbench.log"
find "$R/gpurun_out/pmc" -type f -size +8M -delete
tail -2 "$R/gpurun_out/pmc_kbench.log"
