#!/bin/bash
# Round-2 call 12: full suite with the fixed v4 oracle + v5 default +
# grid-aware dispatch; bench A/B v5 on/off; fresh budget.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call12.log) 2>&1

timeout 1500 python -m pytest tests -m gpu -q 2>&1 | tail -3

for arm in v45 off v45; do
  if [ "$arm" = v45 ]; then unset GORDO_LSTM_V4; else export GORDO_LSTM_V4=0; fi
  timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1 | python -c "
import json,sys
d = json.loads(sys.stdin.read())
print(f'$arm: {d[\"value\"]:.0f} machines/hour ({d[\"ms_per_step\"]:.0f} ms/step)')"
done
unset GORDO_LSTM_V4

timeout 900 python bench.py --gpus 1 --steps 3 --warmup 1 --verbose \
  2>gpurun_out/bench12_verbose.err | tail -1
grep phase_budget gpurun_out/bench12_verbose.err | tail -1
