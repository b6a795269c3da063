#!/bin/bash
# Round-2 second GPU call: big-H dispatch A/B at pack scale, serving
# stage budget + hipGraph + config-#5 shape, bench phase budget, and a
# rocprof capture of the fleet step with the v3 defaults.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call2.log) 2>&1

# 0) new GPU tests only (captured predict, big-H A/B fixtures)
timeout 900 python -m pytest tests/test_gpu_e2e.py tests/test_ops_gpu.py \
  -q -m gpu -x 2>&1 | tail -4

# 1) big-H fused vs per-timestep at PACK scale (G=64, B=256) and small
#    G — decides the dispatch heuristic
timeout 1200 python - <<'PY'
import time, numpy as np, torch
from gordo_amd.engine.pack import LSTMPack
from gordo_amd.engine import pack as packmod
from gordo_amd.engine.spec import LayerSpec, ModelSpec
import gordo_amd.ops as ops

def run(G, batch):
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
    rng = np.random.default_rng(3)
    X = torch.from_numpy(rng.random((G, 600, 50)).astype("float32"))
    p = LSTMPack(spec, G=G, device="cuda", seeds=list(range(G)))
    Xg = X.to("cuda", p.compute_dtype)
    def fit_once(tag):
        t0 = time.perf_counter()
        p.fit(Xg, Xg.clone(), epochs=1, batch_size=batch, shuffle=False)
        torch.cuda.synchronize()
        print(f"G={G:3d} batch={batch} {tag}: {time.perf_counter()-t0:.3f} s/epoch")
    fit_once("warm fused")
    fit_once("FUSED     ")
    avail = ops.lstm_seq_available
    ops.lstm_seq_available = lambda H: False
    packmod.ops.lstm_seq_available = lambda H: False
    fit_once("warm per-t")
    fit_once("PER-T     ")
    ops.lstm_seq_available = avail
    packmod.ops.lstm_seq_available = avail
    del p, Xg
    torch.cuda.empty_cache()

run(16, 256)
run(64, 256)
PY

# 2) serving: stage budget (eager vs hipGraph), HTTP with new fastjson,
#    config-#5 direct shape (10k rows, 16 models, 8 threads)
timeout 600 python scripts/bench_serving.py --rounds 100 --profile-stages 2>/dev/null | tail -1
timeout 600 python scripts/bench_serving.py --rounds 100 --profile-stages --hipgraph 2>/dev/null | tail -1
timeout 600 python scripts/bench_serving.py --rounds 80 --threads 8 --endpoint both 2>/dev/null | tail -1
timeout 600 python scripts/bench_serving.py --rounds 80 --threads 8 --endpoint both --hipgraph 2>/dev/null | tail -1
timeout 600 python scripts/bench_serving.py --rounds 50 --direct --rows 10000 2>/dev/null | tail -1
timeout 900 python scripts/bench_serving.py --rounds 200 --direct --rows 10000 --n-models 16 --threads 8 2>/dev/null | tail -1
timeout 900 python scripts/bench_serving.py --rounds 200 --direct --rows 10000 --n-models 16 --threads 8 --hipgraph 2>/dev/null | tail -1

# 3) bench phase budget with v3 defaults
timeout 1200 python bench.py --gpus 1 --steps 3 --warmup 1 --verbose 2>gpurun_out/bench_verbose.err | tail -1
grep phase_budget gpurun_out/bench_verbose.err | tail -1

# 4) rocprof kernel stats of one fleet step (v3 defaults) -> profiles
export TMPDIR=/tmp
cd /tmp
timeout 900 rocprofv3 --stats -d /tmp/prof_r2 -o fleet_r2 -- \
  python /root/repo/bench.py --gpus 1 --steps 1 --warmup 1 --machines-per-gpu 60 2>&1 | tail -2
cd "$GRAFT_REPO_ROOT" || cd /root/repo
find /tmp/prof_r2 -name "*stats*" -exec cp {} gpurun_out/ \; 2>/dev/null
ls gpurun_out/
