#!/bin/bash
# Round-2 third GPU call: same-box v1-vs-v3 fleet A/B, rocprof kernel
# stats with the current defaults, driver-command dry run, prefork
# HTTP serving on GPU.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call3.log) 2>&1

# 0) quick green check of the full GPU suite
timeout 1200 python -m pytest tests -m gpu -x -q 2>&1 | tail -3

# 1) WITHIN-BOX fleet A/B: v3 default vs GORDO_LSTM_V1=1 (cross-box
#    noise was ~3%; this decides whether v3 actually moves the fleet)
timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
GORDO_LSTM_V1=1 timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1
timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1

# 2) driver-command dry run: the exact torchrun path the driver uses,
#    single rank on this GPU (rank env parsing, device pinning, MAX
#    all-reduce)
timeout 900 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
  --master-addr 127.0.0.1 --master-port 29511 \
  bench.py --gpus 1 --steps 1 --warmup 0 2>&1 | tail -2

# 3) rocprof kernel stats of one fleet step at current defaults
export TMPDIR=/tmp
( cd /tmp && timeout 900 rocprofv3 --kernel-trace --stats \
  -d "$GRAFT_REPO_ROOT/gpurun_out/prof_r2" -- \
  python "$GRAFT_REPO_ROOT/bench.py" --gpus 1 --steps 1 --warmup 1 \
  --machines-per-gpu 60 > "$GRAFT_REPO_ROOT/gpurun_out/prof_stdout.log" 2>&1 )
tail -2 gpurun_out/prof_stdout.log
ls gpurun_out/prof_r2 2>/dev/null | head

# 4) prefork HTTP serving on GPU: 4 and 8 workers, json + parquet note
timeout 900 python scripts/bench_serving.py --rounds 300 --threads 16 \
  --endpoint both --http-workers 4 2>/dev/null | tail -1
timeout 900 python scripts/bench_serving.py --rounds 300 --threads 16 \
  --endpoint both --http-workers 8 2>/dev/null | tail -1
