#!/bin/bash
# Round-2 call 23: validate largest-first group ordering (save_join
# tail) on the final tree; full GPU suite + verbose bench.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call23.log) 2>&1

timeout 400 python -m pytest tests -m gpu -q 2>&1 | tail -2

timeout 240 python bench.py --gpus 1 --steps 3 --warmup 1 --verbose \
  2>gpurun_out/bench23_verbose.err | tail -1
grep phase_budget gpurun_out/bench23_verbose.err | tail -1
