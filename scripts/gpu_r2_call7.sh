#!/bin/bash
# Round-2 seventh GPU call: validate async adopt+save, batched K13
# scoring, fast request decode, exec-prefork serving; re-measure bench.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call7.log) 2>&1

timeout 1500 python -m pytest tests -m gpu -x -q 2>&1 | tail -3

timeout 900 python bench.py --gpus 1 --steps 3 --warmup 1 --verbose \
  2>gpurun_out/bench7_verbose.err | tail -1
grep phase_budget gpurun_out/bench7_verbose.err | tail -1

timeout 600 python scripts/bench_serving.py --rounds 150 --profile-stages 2>/dev/null | tail -1
timeout 900 python scripts/bench_serving.py --rounds 400 --threads 16 \
  --endpoint both --http-workers 4 2>gpurun_out/http7.err | tail -1
tail -2 gpurun_out/http7.err
