#!/bin/bash
# Round-2 call 24: locate the nondeterminism behind the flaky
# v4-vs-twostep geometry (2,64,24,64,128): bitwise-compare repeated
# runs of each arm and print the tolerance margins.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call24.log) 2>&1

timeout 420 python - <<'PY'
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

# arm 1: v4 fused, training + inference, 20 reps bitwise
ref = ops.lstm_seq_fwd_fused(x, Wx, Wh, b, store_aux=True)
nd4 = nd4i = 0
for i in range(20):
    out = ops.lstm_seq_fwd_fused(x, Wx, Wh, b, store_aux=True)
    if not all(torch.equal(a, c) for a, c in zip(ref, out)):
        nd4 += 1
    (hi,) = ops.lstm_seq_fwd_fused(x, Wx, Wh, b, store_aux=False)
    if not torch.equal(hi, ref[0]):
        nd4i += 1
print(f"v4 nondet: train {nd4}/20, inf-vs-train {nd4i}/20")

# arm 2: two-step, 20 reps bitwise
xW = ops.grouped_linear_fwd(x.reshape(G,B*T,F), Wx, b, "linear").view(G,B,T,4*H)
ref1 = ops.lstm_seq_fwd(xW, Wh)
ndx = nd1 = 0
for i in range(20):
    xW2 = ops.grouped_linear_fwd(x.reshape(G,B*T,F), Wx, b, "linear").view(G,B,T,4*H)
    if not torch.equal(xW2, xW):
        ndx += 1
    out1 = ops.lstm_seq_fwd(xW2, Wh)
    if not all(torch.equal(a, c) for a, c in zip(ref1, out1)):
        nd1 += 1
print(f"twostep nondet: xW {ndx}/20, scan {nd1}/20")

# margins against the fp32 oracle
want = ops.lstm_seq_fwd_fused(x.float().cpu(), Wx.float().cpu(),
                              Wh.float().cpu(), b.float().cpu(), store_aux=True)
err4 = (ref[0].float().cpu() - want[0]).abs().mean().item()
err1 = (ref1[0].float().cpu() - want[0]).abs().mean().item()
d = (ref[0].float().cpu() - want[0]).abs()
rel = d / (want[0].abs() + 3e-2)
print(f"err4={err4:.6f} err1={err1:.6f} bound={err1*1.5+0.01:.6f} "
      f"ratio={err4/max(err1,1e-9):.3f}")
print(f"v4-vs-oracle: max_abs={d.max():.4f} frac_over_rtol6e-2={(rel>6e-2).float().mean():.5f}")

# also re-run the pytest node for the authoritative verdict
PY

timeout 300 python -m pytest "tests/test_ops_gpu.py::test_lstm_seq_v4_fused_vs_twostep[2-64-24-64-128]" -q -m gpu 2>&1 | tail -15
