#!/bin/bash
# Round-2 tenth GPU call: green-light the full suite (fixed v4 goldens,
# vectorized wgrad staging), confirm the bench, wgrad staging A/B at
# kernel level, fresh rocprof for profiles/.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call10.log) 2>&1

timeout 1500 python -m pytest tests -m gpu -q 2>&1 | tail -3

timeout 900 python bench.py --gpus 1 --steps 3 --warmup 1 --verbose \
  2>gpurun_out/bench10_verbose.err | tail -1
grep phase_budget gpurun_out/bench10_verbose.err | tail -1

# wgrad staging A/B: vectorized (K,N 8-mult) vs legacy shapes
timeout 600 python - <<'PY'
import time, torch
import gordo_amd.ops as ops
for (M, K, N, tag) in ((36864, 56, 192, "pad8  "), (36864, 50, 168, "legacy")):
    G = 31
    X = torch.randn(G, M, K, device="cuda", dtype=torch.bfloat16)
    dZ = torch.randn(G, M, N, device="cuda", dtype=torch.bfloat16)
    for _ in range(3): ops.grouped_linear_wgrad(X, dZ)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(20): ops.grouped_linear_wgrad(X, dZ)
    torch.cuda.synchronize()
    print(f"wgrad {tag} (M={M} K={K} N={N}): {(time.perf_counter()-t0)/20*1e3:.3f} ms")
PY

# fresh kernel stats at the current defaults
export TMPDIR=/tmp
( cd /tmp && timeout 900 rocprofv3 --kernel-trace --stats \
  -d "$GRAFT_REPO_ROOT/gpurun_out/prof_r2b" -- \
  python "$GRAFT_REPO_ROOT/bench.py" --gpus 1 --steps 1 --warmup 1 \
  --machines-per-gpu 60 > "$GRAFT_REPO_ROOT/gpurun_out/prof_r2b.log" 2>&1 )
tail -1 gpurun_out/prof_r2b.log
find gpurun_out/prof_r2b -name "*.db" | head -1
