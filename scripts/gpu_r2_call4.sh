#!/bin/bash
# Round-2 fourth GPU call: threshold-kernel goldens, bench with the
# async-save + device-threshold path, hipGraph fleet A/B, prefork HTTP
# serving on GPU (proxy-hardened).
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call4.log) 2>&1

# 1) full GPU suite (new K10-K12 goldens included)
timeout 1500 python -m pytest tests -m gpu -x -q 2>&1 | tail -3

# 2) bench with async save + device thresholds (vs call-3's 57.8k)
timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 --verbose \
  2>gpurun_out/bench4_verbose.err | tail -1
grep phase_budget gpurun_out/bench4_verbose.err | tail -1

# 3) hipGraph train-step capture A/B on the same box
GORDO_HIPGRAPH=1 timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1

# 4) prefork HTTP serving on GPU
timeout 900 python scripts/bench_serving.py --rounds 300 --threads 16 \
  --endpoint both --http-workers 4 2>gpurun_out/http4.err | tail -1
timeout 900 python scripts/bench_serving.py --rounds 300 --threads 16 \
  --endpoint both --http-workers 8 2>gpurun_out/http8.err | tail -1
tail -3 gpurun_out/http4.err
