#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
( cd /tmp && PROF_G=31 PROF_B=256 PROF_REPS=2 timeout 600 rocprofv3 \
  --pmc SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_WAVE_CYCLES \
  -d "$GRAFT_REPO_ROOT/gpurun_out/pmc_r2b" -- \
  python "$GRAFT_REPO_ROOT/scripts/prof_lstm.py" \
  > "$GRAFT_REPO_ROOT/gpurun_out/pmc_r2b.log" 2>&1 )
tail -2 gpurun_out/pmc_r2b.log
find gpurun_out/pmc_r2b -name "*.db" | head -1
