#!/bin/bash
# Round-2 ninth GPU call: validate v4 fused-xW scans + A/B on the bench.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call9.log) 2>&1

timeout 1500 python -m pytest tests -m gpu -x -q 2>&1 | tail -3

for arm in v4 twostep v4 twostep; do
  if [ "$arm" = v4 ]; then unset GORDO_LSTM_V4; else export GORDO_LSTM_V4=0; fi
  timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1 2>&1 | tail -1 | python -c "
import json,sys
d = json.loads(sys.stdin.read())
print(f'$arm: {d[\"value\"]:.0f} machines/hour ({d[\"ms_per_step\"]:.0f} ms/step)')"
done
unset GORDO_LSTM_V4

# isolated kernel timing: v4 vs two-step at the bench shape
timeout 600 python - <<'PY'
import time, torch
import gordo_amd.ops as ops
G,B,T,H,F = 31, 256, 144, 48, 56
x = torch.randn(G,B,T,F, device="cuda", dtype=torch.bfloat16)*0.3
Wx = torch.randn(G,F,4*H, device="cuda", dtype=torch.bfloat16)*0.1
Wh = torch.randn(G,H,4*H, device="cuda", dtype=torch.bfloat16)*0.1
b = torch.randn(G,4*H, device="cuda")*0.1
def two():
    xW = ops.grouped_linear_fwd(x.reshape(G,B*T,F), Wx, b, "linear").view(G,B,T,4*H)
    return ops.lstm_seq_fwd(xW, Wh)
def v4():
    return ops.lstm_seq_fwd_fused(x, Wx, Wh, b, True)
def v4inf():
    return ops.lstm_seq_fwd_fused(x, Wx, Wh, b, False)
for name, fn in (("two-step", two), ("v4      ", v4), ("v4-infer", v4inf)):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(15): fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/15*1e3:.3f} ms (G=31 B=256 T=144 H=48 F=56)")
PY
