#!/bin/bash
# Round-2 sixth GPU call: row-tile A/B (16 vs 32 vs 64) on one box —
# call-5's bench regressed 57.9k -> 53.4k after the 16-row default —
# plus a PMC pass locating where the scan cycles go.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call6.log) 2>&1

# 1) row-tile A/B at the bench shape (same box, interleaved)
for rows in 32 16 64 32; do
  GORDO_LSTM_ROWS=$rows timeout 900 python bench.py --gpus 1 --steps 2 \
    --warmup 1 2>&1 | tail -1 | python -c "
import json,sys
d = json.loads(sys.stdin.read())
print(f'rows=$rows: {d[\"value\"]:.0f} machines/hour ({d[\"ms_per_step\"]:.0f} ms/step)')"
done

# 2) scan-kernel A/B at kernel level (isolated shapes)
timeout 600 python - <<'PY'
import os, time, torch
import gordo_amd.ops as ops
for rows in ("32", "16"):
    os.environ["GORDO_LSTM_ROWS"] = rows
    G,B,T,H = 31, 256, 144, 42
    xW = torch.randn(G,B,T,4*H, device="cuda", dtype=torch.bfloat16)
    Wh = torch.randn(G,H,4*H, device="cuda", dtype=torch.bfloat16)*0.1
    for _ in range(3): ops.lstm_seq_fwd(xW, Wh)
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(20): ops.lstm_seq_fwd(xW, Wh)
    torch.cuda.synchronize()
    fwd = (time.perf_counter()-t0)/20*1e3
    hs, cs, ga = ops.lstm_seq_fwd(xW, Wh)
    d = torch.randn(G,B,H, device="cuda", dtype=torch.bfloat16)
    for _ in range(3): ops.lstm_seq_bwd(d, ga, cs, Wh, True)
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(20): ops.lstm_seq_bwd(d, ga, cs, Wh, True)
    torch.cuda.synchronize()
    bwd = (time.perf_counter()-t0)/20*1e3
    print(f"rows={rows}: fwd {fwd:.3f} ms  bwd {bwd:.3f} ms (G=31 B=256)")
PY

# 3) PMC pass: where do the scan cycles go (wait vs issue-stall vs active)
export TMPDIR=/tmp
( cd /tmp && PROF_G=31 PROF_B=256 PROF_REPS=2 timeout 600 rocprofv3 \
  --pmc SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_WAVE_CYCLES \
  -d "$GRAFT_REPO_ROOT/gpurun_out/pmc_r2" -- \
  python "$GRAFT_REPO_ROOT/scripts/prof_lstm.py" \
  > "$GRAFT_REPO_ROOT/gpurun_out/pmc_r2_stdout.log" 2>&1 )
tail -3 gpurun_out/pmc_r2_stdout.log
find gpurun_out/pmc_r2 -name "*.db" | head -2
