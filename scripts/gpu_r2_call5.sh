#!/bin/bash
# Round-2 fifth GPU call: K7/hprev-wgrad validation, bench re-measure,
# exec-prefork HTTP serving, and the recorded sanitize lane.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call5.log) 2>&1

# 1) full GPU suite (window-gather + hprev goldens included)
timeout 1500 python -m pytest tests -m gpu -x -q 2>&1 | tail -3

# 2) bench with K7 gather + hprev wgrad + async saves
timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 --verbose \
  2>gpurun_out/bench5_verbose.err | tail -1
grep phase_budget gpurun_out/bench5_verbose.err | tail -1

# 3) exec-prefork HTTP serving on GPU
timeout 900 python scripts/bench_serving.py --rounds 300 --threads 16 \
  --endpoint both --http-workers 4 2>gpurun_out/http4b.err | tail -1
timeout 900 python scripts/bench_serving.py --rounds 400 --threads 24 \
  --endpoint both --http-workers 8 2>gpurun_out/http8b.err | tail -1
tail -3 gpurun_out/http4b.err

# 4) sanitize lane with recorded output (SURVEY §5.2 evidence)
timeout 1500 bash scripts/gpu_sanitize.sh 2>&1 | tail -6 \
  | tee gpurun_out/sanitize_r02_tail.txt
