#!/bin/bash
# Round-2 first-GPU-call checklist (run via gpurun). Validates everything
# added after round 1's GPU window closed, then re-measures the bench.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo

# 1) full GPU test suite (includes the v3-vs-v1 bit-equality A/B)
timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tail -5

# 2) v3 pipelined LSTM scan A/B at the bench shape
timeout 600 python - <<'PY'
import time, torch
import gordo_amd.ops as ops
torch.cuda.init()
G,B,T,H = 8, 512, 144, 42
xW = torch.randn(G,B,T,4*H, device="cuda", dtype=torch.bfloat16)
Wh = torch.randn(G,H,4*H, device="cuda", dtype=torch.bfloat16)*0.1
for name, fn in (("fwd v1", ops.lstm_seq_fwd), ("fwd v3", ops.lstm_seq_fwd_v3)):
    for _ in range(3): fn(xW, Wh)
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(20): fn(xW, Wh)
    torch.cuda.synchronize()
    print(name, f"{(time.perf_counter()-t0)/20*1e3:.3f} ms")
hs, cs, ga = ops.lstm_seq_fwd(xW, Wh)
dSeq = torch.randn(G,B,H, device="cuda", dtype=torch.bfloat16)
for name, fn in (("bwd v1", ops.lstm_seq_bwd), ("bwd v3", ops.lstm_seq_bwd_v3)):
    for _ in range(3): fn(dSeq, ga, cs, Wh, True)
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(20): fn(dSeq, ga, cs, Wh, True)
    torch.cuda.synchronize()
    print(name, f"{(time.perf_counter()-t0)/20*1e3:.3f} ms")
PY

# 3) bench with the CPU-phase fixes (thresholds O(n), frame fast path)
timeout 1200 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -2

# 4) serving with the C++ JSON encoder, then the codec-free parquet mode
timeout 600 python scripts/bench_serving.py --rounds 60 --threads 8 \
  --endpoint both 2>/dev/null | tail -1
timeout 600 python scripts/bench_serving.py --rounds 60 --threads 8 \
  --endpoint both --format parquet 2>/dev/null | tail -1
timeout 600 python scripts/bench_serving.py --rounds 60 --threads 8 \
  --endpoint both --serve-batch 2>/dev/null | tail -1
