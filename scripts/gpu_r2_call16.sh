#!/bin/bash
# Round-2 call 16: post-revert confirmation + 1000-on-1-GPU bisect.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call16.log) 2>&1

timeout 900 python -m pytest tests/test_ops_gpu.py -m gpu -q 2>&1 | tail -2

# 125-machine step back at the validated state
timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1

# 1000-on-1: steps with a warmup (one-time init/QR/jit excluded) and
# verbose budget to see which phase blew up
timeout 1200 python bench.py --gpus 1 --steps 1 --warmup 1 \
  --machines-per-gpu 1000 --verbose 2>gpurun_out/b1000.err | tail -1
grep phase_budget gpurun_out/b1000.err | tail -1
