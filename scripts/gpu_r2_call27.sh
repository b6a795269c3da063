#!/bin/bash
# Round-2 call 27 (final validation): full GPU suite with the race-fix
# determinism tests, a v5-backward determinism probe, smoke, and the
# final-tree bench.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call27.log) 2>&1

timeout 400 python -m pytest tests -m gpu -q 2>&1 | tail -2

timeout 200 python - <<'PY'
import torch
import gordo_amd.ops as ops

def rand(*s, seed):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(*s, generator=g)

G,B,T,H,F = 2,64,24,64,128
x  = (rand(G,B,T,F, seed=90)*0.5).to("cuda", torch.bfloat16)
Wx = (rand(G,F,4*H, seed=91)*0.2).to("cuda", torch.bfloat16)
Wh = (rand(G,H,4*H, seed=92)*0.2).to("cuda", torch.bfloat16)
b  = (rand(G,4*H, seed=93)*0.1).cuda()
hs, cs, ga = ops.lstm_seq_fwd_fused(x, Wx, Wh, b, store_aux=True)
dSeq = (rand(G,B,T,H, seed=94)*0.1).to("cuda", torch.bfloat16)
ref = ops.lstm_seq_bwd_fused(dSeq, ga, cs, Wh, Wx, last_only=False)
bad = 0
for _ in range(15):
    out = ops.lstm_seq_bwd_fused(dSeq, ga, cs, Wh, Wx, last_only=False)
    if not all(torch.equal(a, c) for a, c in zip(ref, out)):
        bad += 1
print(f"v5 bwd nondet: {bad}/15", flush=True)
PY

timeout 120 python -c "import __graft_entry__ as g; g.smoke(); print('smoke OK')" 2>&1 | tail -1
timeout 240 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
