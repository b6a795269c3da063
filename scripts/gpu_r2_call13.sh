#!/bin/bash
# Round-2 call 13: validate combined wgrad + re-bench; fresh rocprof.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call13.log) 2>&1

timeout 1500 python -m pytest tests -m gpu -q 2>&1 | tail -3

timeout 900 python bench.py --gpus 1 --steps 3 --warmup 1 --verbose \
  2>gpurun_out/bench13_verbose.err | tail -1
grep phase_budget gpurun_out/bench13_verbose.err | tail -1

export TMPDIR=/tmp
( cd /tmp && timeout 900 rocprofv3 --kernel-trace --stats \
  -d "$GRAFT_REPO_ROOT/gpurun_out/prof_r2c" -- \
  python "$GRAFT_REPO_ROOT/bench.py" --gpus 1 --steps 1 --warmup 1 \
  --machines-per-gpu 60 > "$GRAFT_REPO_ROOT/gpurun_out/prof_r2c.log" 2>&1 )
tail -1 gpurun_out/prof_r2c.log
