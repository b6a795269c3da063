#!/usr/bin/env python3
"""Micro-profile of the LSTM pack train_batch: per-op wall times via
torch.cuda events. Run under rocprofv3 --stats for kernel-level data."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from gordo_amd import ops
from gordo_amd.engine.pack import LSTMPack
from gordo_amd.engine.spec import LayerSpec, ModelSpec

G = int(os.environ.get("PROF_G", 62))
B = int(os.environ.get("PROF_B", 256))
T = int(os.environ.get("PROF_T", 144))
F = 50
REPS = int(os.environ.get("PROF_REPS", 5))

spec = ModelSpec(
    model_type="lstm", n_features=F, n_features_out=F,
    layers=[
        LayerSpec(kind="lstm", units=u, return_sequences=(i != 5))
        for i, u in enumerate([42, 33, 25, 25, 33, 42])
    ] + [LayerSpec(kind="dense", units=F, activation="linear")],
    lookback_window=T,
)

dev = "cuda"
pack = LSTMPack(spec, G=G, device=dev, seeds=list(range(G)))
Xw = torch.rand(G, B, T, F, device=dev, dtype=pack.compute_dtype)
Tb = torch.rand(G, B, F, device=dev, dtype=pack.compute_dtype)

print(f"fused path: {pack._use_fused()}", file=sys.stderr)

# warmup
pack.train_batch(Xw, Tb)
torch.cuda.synchronize()

t0 = time.time()
for _ in range(REPS):
    pack.train_batch(Xw, Tb)
torch.cuda.synchronize()
total = (time.time() - t0) / REPS
print(f"train_batch: {total*1000:.1f} ms (G={G}, B={B}, T={T})",
      file=sys.stderr)

# per-op timing of the forward+backward pieces for ONE layer shape
def timeit(fn, reps=10):
    fn()
    torch.cuda.synchronize()
    t = time.time()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t) / reps * 1000


fin, H = F, 42
H4 = 4 * H
Wx = pack.store.cviews["Wx0"]
Wh = pack.store.cviews["Wh0"]
b = pack.store.views["bl0"]
flat = Xw.reshape(G, B * T, fin)
xW = ops.grouped_linear_fwd(flat, Wx, b, "linear")
print(f"xW gemm [G{G},M{B*T},K{fin},N{H4}]: "
      f"{timeit(lambda: ops.grouped_linear_fwd(flat, Wx, b, 'linear')):.2f} ms",
      file=sys.stderr)
xW4 = xW.view(G, B, T, H4)
hs, cs, gacts = ops.lstm_seq_fwd(xW4, Wh)
print(f"lstm_seq_fwd: {timeit(lambda: ops.lstm_seq_fwd(xW4, Wh)):.2f} ms",
      file=sys.stderr)
dSeq = torch.rand_like(hs)
print(f"lstm_seq_bwd: "
      f"{timeit(lambda: ops.lstm_seq_bwd(dSeq, gacts, cs, Wh, False)):.2f} ms",
      file=sys.stderr)
dG = ops.lstm_seq_bwd(dSeq, gacts, cs, Wh, False).view(G, B * T, H4)
print(f"wgrad [G{G},M{B*T},K{fin},N{H4}]: "
      f"{timeit(lambda: ops.grouped_linear_wgrad(flat, dG)):.2f} ms",
      file=sys.stderr)
hflat = hs.reshape(G, B * T, H)
print(f"wgrad Wh [G{G},M{B*T},K{H},N{H4}]: "
      f"{timeit(lambda: ops.grouped_linear_wgrad(hflat, dG)):.2f} ms",
      file=sys.stderr)
print(f"bwd_data dX [G{G},M{B*T},N{H4},K{fin}]: "
      f"{timeit(lambda: ops.grouped_linear_bwd_data(dG, Wx)):.2f} ms",
      file=sys.stderr)
print(f"adam (flat {pack.store.p32.numel()}): "
      f"{timeit(lambda: pack.store.adam_step(1e-3, 0.9, 0.999, 1e-7)):.2f} ms",
      file=sys.stderr)
