#!/bin/bash
# Round-2 eighth GPU call: validate the pad8 layout on GPU and A/B it
# against the unpadded layout on one box.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call8.log) 2>&1

timeout 1500 python -m pytest tests -m gpu -x -q 2>&1 | tail -3

for arm in pad8 nopad pad8 nopad; do
  if [ "$arm" = pad8 ]; then unset GORDO_PAD8; else export GORDO_PAD8=0; fi
  timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1 | python -c "
import json,sys
d = json.loads(sys.stdin.read())
print(f'$arm: {d[\"value\"]:.0f} machines/hour ({d[\"ms_per_step\"]:.0f} ms/step)')"
done
unset GORDO_PAD8

# quick serving spot-check with padded serving packs
timeout 600 python scripts/bench_serving.py --rounds 100 --profile-stages 2>/dev/null | tail -1
timeout 600 python scripts/bench_serving.py --rounds 60 --direct --rows 10000 --n-models 4 --threads 8 2>/dev/null | tail -1
