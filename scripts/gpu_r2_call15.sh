#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call15.log) 2>&1
timeout 1500 python -m pytest tests -m gpu -q 2>&1 | tail -2
timeout 600 python - <<'PY'
import time, torch
import gordo_amd.ops as ops
for (M, K, N, tag) in ((36864, 56, 192, "lstm-Wx+Wh-like"), (36864, 50, 168, "unpadded")):
    G = 31
    X = torch.randn(G, M, K, device="cuda", dtype=torch.bfloat16)
    dZ = torch.randn(G, M, N, device="cuda", dtype=torch.bfloat16)
    for _ in range(3): ops.grouped_linear_wgrad(X, dZ)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(20): ops.grouped_linear_wgrad(X, dZ)
    torch.cuda.synchronize()
    print(f"wgrad {tag} (K={K} N={N}): {(time.perf_counter()-t0)/20*1e3:.3f} ms  (pre-pipeline: 0.254-0.288)")
PY
timeout 900 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
timeout 900 python bench.py --gpus 1 --steps 1 --warmup 0 --machines-per-gpu 1000 2>&1 | tail -1
