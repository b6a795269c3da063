#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call17.log) 2>&1
timeout 1500 python -m pytest tests -m gpu -q 2>&1 | tail -2
timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
timeout 1200 python bench.py --gpus 1 --steps 1 --warmup 1 \
  --machines-per-gpu 1000 --verbose 2>gpurun_out/b1000b.err | tail -1
grep phase_budget gpurun_out/b1000b.err | tail -1
