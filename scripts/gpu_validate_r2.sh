#!/bin/bash
# Round-2 first-GPU-call checklist (run via gpurun). Validates the v3
# pipelined scans, the new big-H (H>64) scan kernels, and re-measures
# the bench + serving with the round-1 CPU-phase fixes in place.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/validate_r2.log) 2>&1

# 1) full GPU test suite (v3-vs-v1 bit-equality A/B + big-H goldens)
timeout 1500 python -m pytest tests -m gpu -x -q 2>&1 | tail -6

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

# 3) big-H (reference default dims 256/128/64) fused vs per-timestep
timeout 900 python - <<'PY'
import time, numpy as np, torch
from gordo_amd.engine.pack import LSTMPack
from gordo_amd.engine import pack as packmod
from gordo_amd.engine.spec import LayerSpec, ModelSpec
import gordo_amd.ops as ops

spec = ModelSpec(
    model_type="lstm", n_features=50, n_features_out=50,
    layers=[
        LayerSpec(kind="lstm", units=256, return_sequences=True),
        LayerSpec(kind="lstm", units=128, return_sequences=True),
        LayerSpec(kind="lstm", units=64, return_sequences=False),
        LayerSpec(kind="dense", units=50, activation="linear"),
    ],
    lookback_window=144,
)
G = 16
rng = np.random.default_rng(3)
X = torch.from_numpy(rng.random((G, 800, 50)).astype("float32"))
p = LSTMPack(spec, G=G, device="cuda", seeds=list(range(G)))
Xg = X.to("cuda", p.compute_dtype)

def fit_once(tag):
    t0 = time.perf_counter()
    p.fit(Xg, Xg.clone(), epochs=1, batch_size=64, shuffle=False)
    torch.cuda.synchronize()
    print(tag, f"{time.perf_counter()-t0:.3f} s/epoch (G={G}, T=144, dims 256/128/64)")

fit_once("warmup   fused")
fit_once("big-H    fused")
fit_once("big-H    fused")
avail = ops.lstm_seq_available
ops.lstm_seq_available = lambda H: False
packmod.ops.lstm_seq_available = lambda H: False
fit_once("warmup   per-t")
fit_once("per-timestep  ")
ops.lstm_seq_available = avail
packmod.ops.lstm_seq_available = avail
PY

# 4) bench with the CPU-phase fixes (thresholds O(n), frame fast path)
timeout 1200 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -2

# 5) serving with the C++ JSON encoder, parquet mode, micro-batcher
timeout 600 python scripts/bench_serving.py --rounds 60 --threads 8 \
  --endpoint both 2>/dev/null | tail -1
timeout 600 python scripts/bench_serving.py --rounds 60 --threads 8 \
  --endpoint both --format parquet 2>/dev/null | tail -1
timeout 600 python scripts/bench_serving.py --rounds 60 --threads 8 \
  --endpoint both --serve-batch 2>/dev/null | tail -1
