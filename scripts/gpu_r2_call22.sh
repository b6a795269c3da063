#!/bin/bash
# Round-2 call 22: A/B the lazy-fetch overlap (GORDO_PREFETCH) on the
# fleet bench; confirm the suite stays green with the restructured
# build_all.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call22.log) 2>&1

timeout 400 python -m pytest tests -m gpu -q 2>&1 | tail -2

for arm in on off on; do
  if [ "$arm" = on ]; then unset GORDO_PREFETCH; else export GORDO_PREFETCH=0; fi
  timeout 240 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1 | python -c "
import json,sys
d = json.loads(sys.stdin.read())
print(f'prefetch-$arm: {d[\"value\"]:.0f} machines/hour ({d[\"ms_per_step\"]:.0f} ms/step)')"
done
unset GORDO_PREFETCH

timeout 240 python bench.py --gpus 1 --steps 3 --warmup 1 --verbose \
  2>gpurun_out/bench22_verbose.err | tail -1
grep phase_budget gpurun_out/bench22_verbose.err | tail -1
