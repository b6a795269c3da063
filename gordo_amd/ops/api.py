"""
Device-op dispatch.

On GPU (ROCm/MI355X) tensors every op runs in the in-tree HIP extension
(``gordo_amd.ops._gordo_hip``, compiled for gfx950 from ``csrc/``); if
the extension is missing on a CUDA-capable host the call FAILS loudly —
there is deliberately no silent eager fallback on GPU. On CPU tensors
the fp32 reference implementations run (the test oracle / CPU lane).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import reference as ref
from .reference import (  # noqa — ACT_* re-exported via gordo_amd.ops
    ACT_LINEAR,
    ACT_TANH,
    ACT_RELU,
    ACT_SIGMOID,
    act_code,
)

_hip = None
_hip_err: Optional[str] = None


def _load_hip():
    global _hip, _hip_err
    if _hip is not None or _hip_err is not None:
        return _hip
    try:
        from . import _gordo_hip  # built in-tree by setup.py / __graft_entry__.build()

        _hip = _gordo_hip
    except ImportError as e:
        _hip_err = str(e)
    return _hip


def hip_available() -> bool:
    return _load_hip() is not None


def _require_hip():
    mod = _load_hip()
    if mod is None:
        raise RuntimeError(
            "gordo_amd HIP extension (_gordo_hip) is not built but a GPU "
            f"tensor was passed. Build it with `python setup.py build_ext "
            f"--inplace` (PYTORCH_ROCM_ARCH=gfx950). Import error: {_hip_err}"
        )
    return mod


def _on_gpu(t: torch.Tensor) -> bool:
    return t.is_cuda


def grouped_linear_fwd(X, W, b, act) -> torch.Tensor:
    act = act_code(act)
    if _on_gpu(X):
        return _require_hip().grouped_linear_fwd(X, W, b, act)
    return ref.grouped_linear_fwd(X, W, b, act)


def act_l1_bwd(dA, Y, act, l1: float) -> torch.Tensor:
    act = act_code(act)
    if _on_gpu(dA):
        return _require_hip().act_l1_bwd(dA, Y, act, float(l1))
    return ref.act_l1_bwd(dA, Y, act, l1)


def grouped_linear_bwd_data(dZ, W) -> torch.Tensor:
    if _on_gpu(dZ):
        return _require_hip().grouped_linear_bwd_data(dZ, W)
    return ref.grouped_linear_bwd_data(dZ, W)


def grouped_linear_wgrad(X, dZ) -> Tuple[torch.Tensor, torch.Tensor]:
    if _on_gpu(X):
        return _require_hip().grouped_linear_wgrad(X, dZ)
    return ref.grouped_linear_wgrad(X, dZ)


def grouped_linear_wgrad_hprev(hs, dZ, T: int):
    """dWh = h_prev^T @ dG over flattened (b, t) rows, where
    h_prev(b, t) = hs[b, t-1] and zero at t == 0 — the in-kernel
    shifted addressing removes the h_prev_all concat from BPTT
    (a full [G,B,T,H] copy per layer per batch)."""
    if _on_gpu(hs):
        return _require_hip().grouped_linear_wgrad_hprev(hs, dZ, int(T))
    G, B, T_, H = hs.shape
    h_prev = torch.cat(
        [torch.zeros_like(hs[:, :, :1]), hs[:, :, :-1]], dim=2
    ).reshape(G, B * T_, H)
    return ref.grouped_linear_wgrad(h_prev, dZ)


def window_gather(X, idx, T: int):
    """K7 sliding-window featurizer: gather [G,B,T,F] lookback windows
    from the resident series [G,N,F] (device analog of
    create_keras_timeseriesgenerator, reference models.py:713-793)."""
    if _on_gpu(X):
        return _require_hip().window_gather(X, idx, int(T))
    G, N, F = X.shape
    B = idx.shape[1]
    rows = idx.unsqueeze(-1).to(torch.long) + torch.arange(T)
    return X.gather(
        1, rows.reshape(G, B * T, 1).expand(G, B * T, F)
    ).view(G, B, T, F)


def grouped_wgrad_xh(seq, hs, dZ, T: int):
    """Combined recurrent weight-grad: (dWx, dWh, db) in one kernel
    pass — dZ is staged once for both GEMMs (it is half of each
    separate call's global traffic)."""
    if _on_gpu(seq):
        out = _require_hip().grouped_wgrad_xh(seq, hs, dZ, int(T))
        return out[0], out[1], out[2]
    dWx, db = ref.grouped_linear_wgrad(seq, dZ)
    dWh, _ = grouped_linear_wgrad_hprev(hs, dZ, T)
    return dWx, dWh, db


def grouped_gemm_acc(A, B, C):
    if _on_gpu(A):
        return _require_hip().grouped_gemm_acc(A, B, C)
    return ref.grouped_gemm_acc(A, B, C)


def mse_bwd(Y, T, real_n: int = -1) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-model MSE loss + grad. ``real_n``: divisor for the mean
    when the feature dim carries zero padding (pad diffs are exactly 0,
    so only the normalization needs the real count)."""
    if _on_gpu(Y):
        return _require_hip().mse_bwd(Y, T, int(real_n))
    return ref.mse_bwd(Y, T, real_n)


def adam_step(p, g, m, v, lr, beta1, beta2, eps, step, p_lp=None,
              step_buf=None):
    if _on_gpu(p):
        if step_buf is None:
            raise ValueError("GPU adam_step needs a device step_buf")
        return _require_hip().adam_step(
            p, g, m, v, float(lr), float(beta1), float(beta2), float(eps),
            int(step), p_lp, step_buf,
        )
    return ref.adam_step(p, g, m, v, lr, beta1, beta2, eps, step, p_lp)


def lstm_seq_fwd(xW, Wh):
    """Fused on-device LSTM sequence scan (GPU only; H<=64 LDS-resident
    kernels, 64<H<=256 streamed-Wh kernels).

    The software-pipelined v3 kernel (double-buffered x-gate register
    prefetch) is the DEFAULT for the barriered H<=64 layout since the
    round-2 GPU A/B (fwd 1.11 vs 1.51 ms, bwd 0.89 vs 0.99 ms at the
    bench shape G=8 B=512 T=144 H=42 — gpurun validate_r2). Set
    ``GORDO_LSTM_V1=1`` to fall back to the unpipelined v1 scan.
    H%16==0 still takes the barrier-free v2 path inside the extension;
    H>64 routes to the big-H kernels in both entry points."""
    import os as _os

    ext = _require_hip()
    G, B = xW.shape[0], xW.shape[1]
    H = xW.shape[-1] // 4
    # v2 (barrier-free, 64-row per-wave blocks) only when its grid
    # fills the chip: at fleet G its G*ceil(B/64) blocks underfill 256
    # CUs and the row-tiled pipelined v3 wins (call-10 rocprof: v2 bwd
    # at 60 WGs averaged 3x the v4 fwd per launch)
    v2_fills = H % 16 == 0 and G * ((B + 63) // 64) >= 256
    if (
        _os.environ.get("GORDO_LSTM_V1") == "1"
        or H > 64            # big-H dispatch lives in the v1 entry
        or v2_fills
    ):
        return ext.lstm_seq_fwd(xW, Wh)
    return ext.lstm_seq_fwd_v3(xW, Wh)


def lstm_v4_available(H: int, F: int) -> bool:
    """v4 fused-xW scan geometry: H <= 64, H%8==0, F%8==0, F<=128
    (the pad8 engine layout satisfies the alignment)."""
    return hip_available() and H <= 64 and H % 8 == 0 and F % 8 == 0 \
        and F <= 128


def lstm_seq_fwd_fused(xseq, Wx, Wh, bias, store_aux=True):
    """v4 scan: the x-side gate GEMM runs INSIDE the recurrence over an
    LDS-resident [Wh ; Wx] tile — no xW round trip through HBM (the
    fleet's scans are bandwidth-bound). ``store_aux=False`` skips the
    cs/gacts stores for inference. GORDO_LSTM_V4=0 falls back to the
    two-step path at the engine level."""
    if _on_gpu(xseq):
        return _require_hip().lstm_seq_fwd_fused(
            xseq, Wx, Wh, bias, bool(store_aux)
        )
    # CPU oracle: explicit xW then the per-timestep reference scan
    G, B, T, F = xseq.shape
    H = Wh.shape[1]
    gates_all = ref.grouped_linear_fwd(
        xseq.reshape(G, B * T, F), Wx, bias, ACT_LINEAR
    ).view(G, B, T, 4 * H)
    h = torch.zeros(G, B, H, dtype=xseq.dtype)
    c = torch.zeros(G, B, H, dtype=torch.float32)
    hs = torch.empty(G, B, T, H, dtype=xseq.dtype)
    cs = torch.empty(G, B, T, H, dtype=torch.float32)
    ga = torch.empty(G, B, T, 4 * H, dtype=xseq.dtype)
    for t in range(T):
        gates = gates_all[:, :, t] + torch.bmm(h, Wh)
        h, c, gact = ref.lstm_pointwise_fwd(gates, c)
        hs[:, :, t] = h
        cs[:, :, t] = c
        ga[:, :, t] = gact
    if store_aux:
        return [hs, cs, ga]
    return [hs]


def lstm_seq_fwd_v3(xW, Wh):
    """The pipelined forward scan, directly (for A/B tests)."""
    return _require_hip().lstm_seq_fwd_v3(xW, Wh)


def lstm_seq_bwd(dSeq, gacts, cs, Wh, last_only):
    """Fused on-device LSTM backward (BPTT) scan (GPU only). Same
    version dispatch as ``lstm_seq_fwd``: pipelined v3 by default for
    the barriered layout, v2 for H%16==0, big-H kernels for H>64;
    ``GORDO_LSTM_V1=1`` forces the unpipelined v1 scan."""
    import os as _os

    ext = _require_hip()
    G, B = gacts.shape[0], gacts.shape[1]
    H = gacts.shape[-1] // 4
    v2_fills = H % 16 == 0 and G * ((B + 63) // 64) >= 256
    if (
        _os.environ.get("GORDO_LSTM_V1") == "1"
        or H > 64
        or v2_fills
    ):
        return ext.lstm_seq_bwd(dSeq, gacts, cs, Wh, last_only)
    return ext.lstm_seq_bwd_v3(dSeq, gacts, cs, Wh, last_only)


def lstm_seq_bwd_fused(dSeq, gacts, cs, Wh, Wx, last_only):
    """v5 reverse scan with the bwd-data GEMM fused in: returns
    (dG, dX) where dX = dG @ Wx^T computed per step against an
    LDS-resident WxT (no dG HBM re-read). Geometry: H<=64, F<=128."""
    if _on_gpu(gacts):
        out = _require_hip().lstm_seq_bwd_fused(
            dSeq, gacts, cs, Wh, Wx, bool(last_only)
        )
        return out[0], out[1]
    dG = lstm_seq_bwd(dSeq, gacts, cs, Wh, last_only)
    G, B, T, H4 = gacts.shape
    dX = ref.grouped_linear_bwd_data(
        dG.view(G, B * T, H4), Wx
    ).view(G, B, T, Wx.shape[1])
    return dG, dX


def lstm_seq_bwd_v3(dSeq, gacts, cs, Wh, last_only):
    """The pipelined backward scan, directly (for A/B tests)."""
    return _require_hip().lstm_seq_bwd_v3(dSeq, gacts, cs, Wh, last_only)


def lstm_seq_available(H: int) -> bool:
    """Fused sequence-scan coverage: the LDS-resident kernels serve
    H <= 64; the big-H kernels (Wh streamed from L2, 64-column tile
    loop) serve 64 < H <= 256 when H % 8 == 0 — which covers the
    reference's default LSTM dims (256, 128, 64), reference
    gordo/machine/model/factories/lstm_autoencoder.py:112."""
    if not hip_available():
        return False
    return H <= 64 or (H <= 256 and H % 8 == 0)


def anomaly_score(out, y, scale, minv, feat_thr, agg_thr):
    """Fused serving-path anomaly scoring (GPU only; callers fall back
    to numpy/pandas on CPU — machine/model/anomaly/diff.py)."""
    return _require_hip().anomaly_score(out, y, scale, minv, feat_thr,
                                        float(agg_thr))


def trail_min_max(X, w: int):
    """Per-row ``rolling(w).min().max()`` (K10). GPU tensors run the
    HIP kernel; CPU falls back to the scipy O(n) path
    (machine/model/utils.trail_min_max)."""
    if _on_gpu(X):
        return _require_hip().trail_min_max(X, int(w))
    from ..machine.model.utils import trail_min_max as cpu_tmm
    import numpy as _np

    return torch.as_tensor(
        _np.atleast_1d(cpu_tmm(X.numpy().T, int(w))), dtype=torch.float32
    )


def windowed_quantile(X, w: int, q: float):
    """Per-row rolling(w).quantile(q) with pandas min_periods=w NaN
    semantics (K11; q=0.5 == the smm rolling median)."""
    if _on_gpu(X):
        return _require_hip().windowed_quantile(X, int(w), float(q))
    import pandas as _pd

    df = _pd.DataFrame(X.numpy().T)
    out = df.rolling(int(w)).quantile(float(q)).to_numpy().T[:, int(w) - 1:]
    return torch.as_tensor(out, dtype=torch.float32)


def row_quantile(X, q: float):
    """Per-row NaN-dropping linear-interpolated quantile (K12)."""
    if _on_gpu(X):
        return _require_hip().row_quantile(X, float(q))
    import numpy as _np

    return torch.as_tensor(
        _np.nanquantile(X.numpy(), float(q), axis=1), dtype=torch.float32
    )


def lstm_pointwise_fwd(gates, c_prev):
    if _on_gpu(gates):
        return _require_hip().lstm_pointwise_fwd(gates, c_prev)
    return ref.lstm_pointwise_fwd(gates, c_prev)


def lstm_pointwise_bwd(dh, dc_next, gact, c, c_prev):
    if _on_gpu(dh):
        return _require_hip().lstm_pointwise_bwd(dh, dc_next, gact, c, c_prev)
    return ref.lstm_pointwise_bwd(dh, dc_next, gact, c, c_prev)
