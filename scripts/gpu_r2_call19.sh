#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call19.log) 2>&1
for cap in 256 128; do
  GORDO_MAX_PACK_LSTM=$cap GORDO_MAX_PACK_DENSE=1024 timeout 1200 \
    python bench.py --gpus 1 --steps 1 --warmup 1 --machines-per-gpu 1000 \
    2>&1 | tail -1 | python -c "
import json,sys
d = json.loads(sys.stdin.read())
print(f'cap=$cap: {d[\"value\"]:.0f} machines/hour ({d[\"ms_per_step\"]/1000:.1f} s/step)')"
done
