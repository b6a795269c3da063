#!/bin/bash
# Round-2 call 20: near-final validation — suite, smoke, bench, serving.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call20.log) 2>&1
timeout 1500 python -m pytest tests -m gpu -q 2>&1 | tail -2
timeout 600 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1
timeout 900 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
timeout 600 python scripts/bench_serving.py --rounds 200 --threads 16 \
  --endpoint both --http-workers 4 2>/dev/null | tail -1
timeout 600 python scripts/bench_serving.py --rounds 60 --direct \
  --rows 10000 --n-models 16 --threads 8 --hipgraph 2>/dev/null | tail -1
