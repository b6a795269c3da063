#!/bin/bash
# Round-2 call 21 (final): full GPU suite + smoke + bench on the final
# tree (defensive _pad_io, fold caps, init-cache widening, key hoisting).
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call21.log) 2>&1

timeout 560 python -m pytest tests -m gpu -q 2>&1 | tail -3

timeout 120 python -c "import __graft_entry__ as g; g.smoke(); print('smoke OK')"

timeout 240 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
