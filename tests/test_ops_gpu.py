"""HIP kernel numerics vs the CPU fp32 reference (gordo_amd.ops.reference).

Every op is compared against the plain PyTorch fp32 oracle at bf16
tolerances. Shapes cover the framework's hot cases: odd feature counts
(50 tags, hourglass dims 38/28/19), edge tiles, and LSTM gate widths.
"""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from gordo_amd import ops
from gordo_amd.ops import reference as ref


def require_hip():
    assert ops.hip_available(), (
        "HIP extension not built — GPU tests must not fall back to eager"
    )


def _rand(*shape, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.rand(*shape, generator=g) * 2 - 1


def to_dev_bf16(t):
    return t.to("cuda", torch.bfloat16)


# bf16 has ~3 decimal digits; accumulation in fp32.
RTOL, ATOL = 2.5e-2, 2.5e-2


@pytest.mark.parametrize(
    "G,M,K,N",
    [
        (1, 64, 64, 64),
        (3, 100, 50, 38),     # 50-tag hourglass first layer
        (5, 33, 19, 28),      # odd everything, edge tiles
        (2, 256, 40, 160),    # LSTM gates shape (4H=160)
        (1, 4176, 50, 152),   # big-M LSTM x-side GEMM
    ],
)
@pytest.mark.parametrize("act", ["linear", "tanh", "sigmoid", "relu"])
def test_grouped_linear_fwd(G, M, K, N, act):
    require_hip()
    X, W = _rand(G, M, K, seed=1), _rand(G, K, N, seed=2)
    b = _rand(G, N, seed=3)
    want = ref.grouped_linear_fwd(X, W, b, ref.act_code(act))
    got = ops.grouped_linear_fwd(
        to_dev_bf16(X), to_dev_bf16(W), b.cuda(), act
    ).float().cpu()
    torch.testing.assert_close(got, want, rtol=RTOL, atol=ATOL)


def test_gemm_not_transposed():
    """Asymmetric operand check (guide §3): catches silent C^T."""
    require_hip()
    G, M, K, N = 1, 32, 32, 48
    X = torch.zeros(G, M, K)
    for i in range(M):
        X[0, i, i % K] = 1.0  # permuted identity-ish
    W = torch.arange(K * N, dtype=torch.float32).view(G, K, N) / (K * N)
    b = torch.zeros(G, N)
    want = ref.grouped_linear_fwd(X, W, b, 0)
    got = ops.grouped_linear_fwd(
        to_dev_bf16(X), to_dev_bf16(W), b.cuda(), 0
    ).float().cpu()
    torch.testing.assert_close(got, want, rtol=RTOL, atol=ATOL)


@pytest.mark.parametrize(
    "G,M,N,K", [(2, 100, 38, 50), (3, 65, 17, 33), (1, 1000, 152, 40)]
)
def test_grouped_linear_bwd_data(G, M, N, K):
    require_hip()
    dZ, W = _rand(G, M, N, seed=4), _rand(G, K, N, seed=5)
    want = ref.grouped_linear_bwd_data(dZ, W)
    got = ops.grouped_linear_bwd_data(
        to_dev_bf16(dZ), to_dev_bf16(W)
    ).float().cpu()
    # inner dim is N here; bf16 input rounding accumulates ~ sqrt(N)
    atol = ATOL * max(1.0, (N / 64) ** 0.5)
    torch.testing.assert_close(got, want, rtol=5e-2, atol=atol)


@pytest.mark.parametrize(
    "G,M,K,N", [
        (2, 200, 50, 38),      # legacy scalar staging (K,N not 8-mult)
        (1, 4176, 40, 160),    # vectorized A+Z staging (pad8 layout)
        (4, 37, 19, 21),       # ragged tails, scalar path
        (2, 333, 56, 48),      # vectorized with ragged M tail
        (1, 512, 48, 21),      # vecA + legacy Z mix
    ]
)
def test_grouped_linear_wgrad(G, M, K, N):
    require_hip()
    X, dZ = _rand(G, M, K, seed=6), _rand(G, M, N, seed=7) * 0.1
    wantW, wantb = ref.grouped_linear_wgrad(X, dZ)
    gotW, gotb = ops.grouped_linear_wgrad(to_dev_bf16(X), to_dev_bf16(dZ))
    # fp32 accumulation over large M: scale tolerance with sqrt(M)
    tol = dict(rtol=3e-2, atol=2e-2 * max(1.0, (M / 256) ** 0.5))
    torch.testing.assert_close(gotW.cpu(), wantW, **tol)
    torch.testing.assert_close(gotb.cpu(), wantb, **tol)


def test_grouped_gemm_acc():
    require_hip()
    G, M, K, N = 3, 64, 24, 96
    A, B = _rand(G, M, K, seed=8), _rand(G, K, N, seed=9)
    C0 = _rand(G, M, N, seed=10)
    want = C0.clone()
    ref.grouped_gemm_acc(A, B, want)
    got = to_dev_bf16(C0).contiguous()
    ops.grouped_gemm_acc(to_dev_bf16(A), to_dev_bf16(B), got)
    torch.testing.assert_close(got.float().cpu(), want, rtol=3e-2, atol=3e-2)


@pytest.mark.parametrize("act,l1", [("tanh", 0.0), ("tanh", 1e-4),
                                    ("linear", 1e-4), ("sigmoid", 0.0)])
def test_act_l1_bwd(act, l1):
    require_hip()
    dA, Y = _rand(2, 100, 30, seed=11), _rand(2, 100, 30, seed=12)
    want = ref.act_l1_bwd(dA, Y, ref.act_code(act), l1)
    got = ops.act_l1_bwd(to_dev_bf16(dA), to_dev_bf16(Y), act, l1)
    torch.testing.assert_close(got.float().cpu(), want, rtol=3e-2, atol=1e-2)


def test_mse_bwd():
    require_hip()
    Y, T = _rand(3, 500, 50, seed=13), _rand(3, 500, 50, seed=14)
    want_loss, want_dY = ref.mse_bwd(Y, T)
    got_loss, got_dY = ops.mse_bwd(to_dev_bf16(Y), to_dev_bf16(T))
    torch.testing.assert_close(got_loss.cpu(), want_loss, rtol=1e-2, atol=1e-3)
    torch.testing.assert_close(
        got_dY.float().cpu(), want_dY, rtol=2e-2, atol=1e-4
    )


def test_adam_step():
    require_hip()
    n = 10007
    p = _rand(n, seed=15)
    g = _rand(n, seed=16)
    m = torch.zeros(n)
    v = torch.zeros(n)
    pd, gd, md, vd = (t.cuda() for t in (p, g, m, v))
    plp = pd.to(torch.bfloat16)
    step_buf = torch.zeros(1, dtype=torch.int32, device="cuda")
    for step in (1, 2, 3):
        ref.adam_step(p, g, m, v, 0.01, 0.9, 0.999, 1e-7, step)
        ops.adam_step(pd, gd, md, vd, 0.01, 0.9, 0.999, 1e-7, step, plp,
                      step_buf=step_buf)
    torch.testing.assert_close(pd.cpu(), p, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(md.cpu(), m, rtol=1e-5, atol=1e-7)
    torch.testing.assert_close(plp.float().cpu(), p, rtol=1e-2, atol=1e-2)


def test_lstm_pointwise_roundtrip():
    require_hip()
    G, B, H = 2, 64, 38
    gates = _rand(G, B, 4 * H, seed=17) * 2
    c_prev = _rand(G, B, H, seed=18)
    want_h, want_c, want_gact = ref.lstm_pointwise_fwd(gates, c_prev)
    got_h, got_c, got_gact = ops.lstm_pointwise_fwd(
        to_dev_bf16(gates), c_prev.cuda()
    )
    torch.testing.assert_close(got_h.float().cpu(), want_h, rtol=2e-2, atol=1e-2)
    torch.testing.assert_close(got_c.cpu(), want_c, rtol=2e-2, atol=1e-2)

    dh = _rand(G, B, H, seed=19)
    dc_next = _rand(G, B, H, seed=20) * 0.1
    want_dg, want_dcp = ref.lstm_pointwise_bwd(
        dh, dc_next, want_gact, want_c, c_prev
    )
    got_dg, got_dcp = ops.lstm_pointwise_bwd(
        to_dev_bf16(dh), dc_next.cuda(), got_gact, got_c, c_prev.cuda()
    )
    torch.testing.assert_close(
        got_dg.float().cpu(), want_dg, rtol=3e-2, atol=2e-2
    )
    torch.testing.assert_close(got_dcp.cpu(), want_dcp, rtol=3e-2, atol=2e-2)


@pytest.mark.parametrize("G,B,T,H,last_only", [
    (2, 64, 12, 38, False),
    (1, 100, 20, 42, True),   # edge row tile, odd H
    (3, 32, 144, 25, True),   # full lookback
])
def test_lstm_seq_fused_vs_reference(G, B, T, H, last_only):
    _check_lstm_seq(G, B, T, H, last_only, wh_scale=0.3)


@pytest.mark.parametrize("G,B,T,H,last_only", [
    (2, 48, 12, 72, False),    # 64-row tile path (B >= 48)
    (1, 33, 20, 128, True),    # ragged row tile, reference default dim
    (2, 64, 24, 256, True),    # largest supported H
    (1, 20, 16, 96, False),    # 32-row tile path (B < 48)
])
def test_lstm_seq_big_vs_reference(G, B, T, H, last_only):
    """Big-H (64 < H <= 256) scan kernels — Wh streamed from L2 —
    vs the per-timestep fp32 oracle (reference default LSTM dims are
    256/128/64: lstm_autoencoder.py:112)."""
    _check_lstm_seq(G, B, T, H, last_only, wh_scale=2.4 / (H ** 0.5))


def _check_lstm_seq(G, B, T, H, last_only, wh_scale):
    """Fused sequence-scan kernels vs the per-timestep fp32 oracle."""
    require_hip()
    H4 = 4 * H
    xW = _rand(G, B, T, H4, seed=30)
    Wh = _rand(G, H, H4, seed=31) * wh_scale

    # reference forward (fp32, per-timestep)
    hs_ref = torch.empty(G, B, T, H)
    cs_ref = torch.empty(G, B, T, H)
    ga_ref = torch.empty(G, B, T, H4)
    h = torch.zeros(G, B, H)
    c = torch.zeros(G, B, H)
    for t in range(T):
        gates = xW[:, :, t] + torch.bmm(h, Wh)
        h, c, ga = ref.lstm_pointwise_fwd(gates, c)
        hs_ref[:, :, t] = h
        cs_ref[:, :, t] = c
        ga_ref[:, :, t] = ga

    got_hs, got_cs, got_ga = ops.lstm_seq_fwd(to_dev_bf16(xW), to_dev_bf16(Wh))
    torch.testing.assert_close(
        got_hs.float().cpu(), hs_ref, rtol=5e-2, atol=3e-2
    )
    torch.testing.assert_close(got_cs.cpu(), cs_ref, rtol=5e-2, atol=3e-2)

    # reference backward
    if last_only:
        dSeq = _rand(G, B, H, seed=32)
    else:
        dSeq = _rand(G, B, T, H, seed=32)
    dG_ref = torch.empty(G, B, T, H4)
    dh = torch.zeros(G, B, H)
    dc = torch.zeros(G, B, H)
    for t in range(T - 1, -1, -1):
        dh_t = dh.clone()
        if last_only:
            if t == T - 1:
                dh_t += dSeq
        else:
            dh_t += dSeq[:, :, t]
        c_prev = cs_ref[:, :, t - 1] if t > 0 else torch.zeros(G, B, H)
        dgates, dc = ref.lstm_pointwise_bwd(
            dh_t, dc, ga_ref[:, :, t], cs_ref[:, :, t], c_prev
        )
        dG_ref[:, :, t] = dgates
        dh = torch.bmm(dgates, Wh.transpose(1, 2))

    got_dG = ops.lstm_seq_bwd(
        to_dev_bf16(dSeq), got_ga, got_cs, to_dev_bf16(Wh), last_only
    )
    torch.testing.assert_close(
        got_dG.float().cpu(), dG_ref, rtol=8e-2, atol=4e-2
    )


def test_anomaly_score_kernel_vs_pandas():
    """K9 fused serving kernel vs the exact pandas/numpy formulas."""
    require_hip()
    import numpy as np
    from sklearn.preprocessing import MinMaxScaler

    rng = np.random.default_rng(5)
    N, F = 1000, 50
    out = rng.random((N, F)).astype("float32")
    y = rng.random((N, F)).astype("float32")
    scaler = MinMaxScaler().fit(rng.random((200, F)) * 3)
    thr = rng.random(F).astype("float32") + 0.1
    agg = 0.37

    ts_ref = np.abs(scaler.transform(out) - scaler.transform(y))
    tots_ref = (ts_ref ** 2).mean(axis=1)
    tu_ref = np.abs(out - y)
    totu_ref = (tu_ref ** 2).mean(axis=1)

    ts, tots, tu, totu, conf, tconf = ops.anomaly_score(
        torch.as_tensor(out, device="cuda"),
        torch.as_tensor(y, device="cuda"),
        torch.as_tensor(scaler.scale_.astype("float32"), device="cuda"),
        torch.as_tensor(scaler.min_.astype("float32"), device="cuda"),
        torch.as_tensor(thr, device="cuda"),
        agg,
    )
    np.testing.assert_allclose(ts.cpu().numpy(), ts_ref, rtol=1e-5, atol=1e-6)
    np.testing.assert_allclose(tots.cpu().numpy(), tots_ref, rtol=1e-5,
                               atol=1e-6)
    np.testing.assert_allclose(tu.cpu().numpy(), tu_ref, rtol=1e-6, atol=1e-7)
    np.testing.assert_allclose(totu.cpu().numpy(), totu_ref, rtol=1e-5,
                               atol=1e-7)
    np.testing.assert_allclose(conf.cpu().numpy(), tu_ref / thr, rtol=1e-5,
                               atol=1e-6)
    np.testing.assert_allclose(tconf.cpu().numpy(), tots_ref / agg,
                               rtol=1e-5, atol=1e-6)


def test_anomaly_frame_device_path_matches_pandas():
    """DiffBased.anomaly at >=512 rows takes the fused kernel path; the
    frame must equal the pandas path bitwise-close."""
    require_hip()
    import numpy as np
    import pandas as pd
    from sklearn.pipeline import Pipeline
    from sklearn.preprocessing import MinMaxScaler

    from gordo_amd.machine.model.anomaly.diff import DiffBasedAnomalyDetector
    from gordo_amd.machine.model.models import KerasAutoEncoder

    rng = np.random.default_rng(6)
    X = pd.DataFrame(rng.random((800, 10)),
                     columns=[f"t{i}" for i in range(10)])
    det = DiffBasedAnomalyDetector(
        base_estimator=Pipeline([
            ("mms", MinMaxScaler()),
            ("ae", KerasAutoEncoder(kind="feedforward_hourglass", epochs=1)),
        ]),
        require_thresholds=False,
    )
    det.fit(X, X)
    frame_gpu = det.anomaly(X, X)
    # force the pandas path and compare
    det._DEVICE_SCORE_MIN_ROWS = 10 ** 9
    frame_cpu = det.anomaly(X, X)
    for col in ("tag-anomaly-scaled", "total-anomaly-scaled",
                "tag-anomaly-unscaled", "total-anomaly-unscaled"):
        np.testing.assert_allclose(
            np.asarray(frame_gpu[col], dtype=float),
            np.asarray(frame_cpu[col], dtype=float),
            rtol=1e-4, atol=1e-5,
        )


@pytest.mark.parametrize("G,B,T,H,last_only", [
    (2, 64, 12, 16, False),
    (1, 100, 20, 48, True),    # Hp=48 (the padded hourglass-42 case)
    (2, 30, 144, 32, True),    # full lookback, ragged rows
    (1, 64, 10, 64, False),    # HF=4
])
def test_lstm_seq_v2_vs_reference(G, B, T, H, last_only):
    """The v2 (register-resident, H%16==0) scan kernels vs the
    per-timestep fp32 oracle — same harness as the v1 test."""
    require_hip()
    H4 = 4 * H
    xW = _rand(G, B, T, H4, seed=40)
    Wh = _rand(G, H, H4, seed=41) * 0.3

    hs_ref = torch.empty(G, B, T, H)
    cs_ref = torch.empty(G, B, T, H)
    ga_ref = torch.empty(G, B, T, H4)
    h = torch.zeros(G, B, H)
    c = torch.zeros(G, B, H)
    for t in range(T):
        gates = xW[:, :, t] + torch.bmm(h, Wh)
        h, c, ga = ref.lstm_pointwise_fwd(gates, c)
        hs_ref[:, :, t] = h
        cs_ref[:, :, t] = c
        ga_ref[:, :, t] = ga

    got_hs, got_cs, got_ga = ops.lstm_seq_fwd(to_dev_bf16(xW), to_dev_bf16(Wh))
    torch.testing.assert_close(got_hs.float().cpu(), hs_ref, rtol=5e-2,
                               atol=3e-2)
    torch.testing.assert_close(got_cs.cpu(), cs_ref, rtol=5e-2, atol=3e-2)

    dSeq = _rand(G, B, H, seed=42) if last_only else _rand(G, B, T, H, seed=42)
    dG_ref = torch.empty(G, B, T, H4)
    dh = torch.zeros(G, B, H)
    dc = torch.zeros(G, B, H)
    for t in range(T - 1, -1, -1):
        dh_t = dh.clone()
        if last_only:
            if t == T - 1:
                dh_t += dSeq
        else:
            dh_t += dSeq[:, :, t]
        c_prev = cs_ref[:, :, t - 1] if t > 0 else torch.zeros(G, B, H)
        dgates, dc = ref.lstm_pointwise_bwd(dh_t, dc, ga_ref[:, :, t],
                                            cs_ref[:, :, t], c_prev)
        dG_ref[:, :, t] = dgates
        dh = torch.bmm(dgates, Wh.transpose(1, 2))

    got_dG = ops.lstm_seq_bwd(to_dev_bf16(dSeq), got_ga, got_cs,
                              to_dev_bf16(Wh), last_only)
    torch.testing.assert_close(got_dG.float().cpu(), dG_ref, rtol=8e-2,
                               atol=4e-2)


@pytest.mark.gpu
@pytest.mark.parametrize("G,B,T,H", [
    (2, 64, 12, 38),
    (1, 100, 20, 42),    # edge row tile, odd H
    (3, 32, 144, 25),    # full lookback
    (4, 130, 50, 60),    # near the H cap, ragged rows (64 would hit
                          # the v2 dispatch in lstm_seq_fwd: H%16==0)
])
def test_lstm_seq_v3_matches_v1(G, B, T, H, monkeypatch):
    """The pipelined v3 forward scan (the default since round 2) is
    numerically identical to the v1 kernel — same layout, same math,
    only the x-gate tile is double-buffered. Exact equality expected
    (bit-identical inputs, same op order). GORDO_LSTM_V1 forces the
    v1 arm through the public dispatch."""
    require_hip()
    H4 = 4 * H
    xW = to_dev_bf16(_rand(G, B, T, H4, seed=40))
    Wh = to_dev_bf16(_rand(G, H, H4, seed=41) * 0.3)
    monkeypatch.setenv("GORDO_LSTM_V1", "1")
    v1_hs, v1_cs, v1_ga = ops.lstm_seq_fwd(xW, Wh)
    monkeypatch.delenv("GORDO_LSTM_V1")
    v3_hs, v3_cs, v3_ga = ops.lstm_seq_fwd_v3(xW, Wh)
    assert torch.equal(v1_hs, v3_hs)
    assert torch.equal(v1_cs, v3_cs)
    assert torch.equal(v1_ga, v3_ga)


@pytest.mark.gpu
@pytest.mark.parametrize("G,B,T,H,last_only", [
    (2, 64, 12, 38, False),
    (1, 100, 20, 42, True),
    (3, 32, 144, 25, True),
    (4, 130, 50, 60, False),
])
def test_lstm_seq_bwd_v3_matches_v1(G, B, T, H, last_only, monkeypatch):
    """The pipelined v3 backward scan is bit-identical to v1 — same
    math, the gate/cell loads are just double-buffered."""
    require_hip()
    H4 = 4 * H
    xW = to_dev_bf16(_rand(G, B, T, H4, seed=50))
    Wh = to_dev_bf16(_rand(G, H, H4, seed=51) * 0.3)
    hs, cs, ga = ops.lstm_seq_fwd(xW, Wh)
    if last_only:
        dSeq = to_dev_bf16(_rand(G, B, H, seed=52))
    else:
        dSeq = to_dev_bf16(_rand(G, B, T, H, seed=52))
    monkeypatch.setenv("GORDO_LSTM_V1", "1")
    v1 = ops.lstm_seq_bwd(dSeq, ga, cs, Wh, last_only)
    monkeypatch.delenv("GORDO_LSTM_V1")
    v3 = ops.lstm_seq_bwd_v3(dSeq, ga, cs, Wh, last_only)
    assert torch.equal(v1, v3)


def test_trail_min_max_kernel_vs_pandas():
    """K10 device kernel vs pandas rolling(w).min().max()."""
    require_hip()
    import pandas as pd

    rng = np.random.default_rng(5)
    X = rng.random((37, 4320)).astype("float32") * 10
    for w in (6, 144):
        got = ops.trail_min_max(torch.tensor(X, device="cuda"), w)
        want = pd.DataFrame(X.T).rolling(w).min().max(axis=0).to_numpy()
        np.testing.assert_allclose(
            got.cpu().numpy(), want, rtol=1e-6, atol=1e-6
        )
    # short series -> NaN (pandas: all-NaN rolling -> max NaN)
    short = torch.tensor(X[:2, :4], device="cuda")
    assert torch.isnan(ops.trail_min_max(short, 6)).all()


def test_windowed_quantile_kernel_vs_pandas():
    """K11 device kernel (q=0.5 == pandas rolling median,
    min_periods=w NaN semantics)."""
    require_hip()
    import pandas as pd

    rng = np.random.default_rng(6)
    X = rng.random((9, 700)).astype("float32")
    X[3, 100:110] = np.nan  # NaN patch -> NaN windows
    for w, q in ((144, 0.5), (7, 0.5), (12, 0.25)):
        got = ops.windowed_quantile(
            torch.tensor(X, device="cuda"), w, q
        ).cpu().numpy()
        want = (
            pd.DataFrame(X.T).rolling(w).quantile(q)
            .to_numpy().T[:, w - 1:]
        )
        np.testing.assert_allclose(got, want, rtol=2e-6, atol=2e-6)


def test_row_quantile_kernel_vs_numpy():
    """K12 device kernel vs NaN-dropping linear-interp quantile."""
    require_hip()
    rng = np.random.default_rng(7)
    X = rng.random((21, 4177)).astype("float32")
    X[2, :50] = np.nan
    X[20, ::3] = np.nan
    for q in (0.99, 0.5, 0.1):
        got = ops.row_quantile(torch.tensor(X, device="cuda"), q)
        want = np.nanquantile(X.astype("float64"), q, axis=1)
        np.testing.assert_allclose(
            got.cpu().numpy(), want, rtol=2e-5, atol=2e-6
        )


def test_device_fold_thresholds_match_cpu():
    """The batched device threshold path of the packed builder (K10
    over scaled-MSE/MAE with per-machine MinMax scaling) equals the
    CPU pandas/scipy computation."""
    require_hip()
    from sklearn.preprocessing import MinMaxScaler

    from gordo_amd.machine.model.utils import trail_min_max as cpu_tmm

    rng = np.random.default_rng(8)
    G, n_tr, n_te, F = 5, 400, 150, 7
    ytr = rng.random((G, n_tr, F)).astype("float32")
    yte = rng.random((G, n_te, F)).astype("float32")
    pred = rng.random((G, n_te, F)).astype("float32")

    ytr_d = torch.tensor(ytr, device="cuda")
    yte_d = torch.tensor(yte, device="cuda").float()
    pred_d = torch.tensor(pred, device="cuda").float()
    ymin = ytr_d.amin(dim=1)
    yrange = ytr_d.amax(dim=1) - ymin
    scale = torch.where(
        yrange == 0, torch.ones_like(yrange), 1.0 / yrange
    ).unsqueeze(1)
    scaled_mse = (((pred_d - yte_d) * scale) ** 2).mean(dim=2)
    mae = (yte_d - pred_d).abs()
    agg = ops.trail_min_max(scaled_mse, 6).cpu().numpy()
    tag = (
        ops.trail_min_max(
            mae.transpose(1, 2).reshape(G * F, n_te).contiguous(), 6
        ).cpu().numpy().reshape(G, F)
    )
    for g in range(G):
        sc = MinMaxScaler().fit(ytr[g])
        sm = ((sc.transform(pred[g]) - sc.transform(yte[g])) ** 2).mean(
            axis=1
        )
        np.testing.assert_allclose(agg[g], cpu_tmm(sm, 6), rtol=1e-4)
        np.testing.assert_allclose(
            tag[g], cpu_tmm(np.abs(yte[g] - pred[g]), 6), rtol=1e-4
        )


@pytest.mark.parametrize("G,B,T,H,N4", [(3, 16, 12, 42, 168),
                                        (1, 9, 7, 25, 100)])
def test_grouped_wgrad_hprev_matches_concat(G, B, T, H, N4):
    """The shifted-addressing recurrent wgrad (no h_prev_all concat)
    equals wgrad over the explicitly concatenated h_prev rows."""
    require_hip()
    hs = _rand(G, B, T, H, seed=60)
    dG = _rand(G, B * T, N4, seed=61)
    h_prev = torch.cat(
        [torch.zeros_like(hs[:, :, :1]), hs[:, :, :-1]], dim=2
    ).reshape(G, B * T, H)
    want_W, want_b = ops.grouped_linear_wgrad(
        to_dev_bf16(h_prev), to_dev_bf16(dG)
    )
    got_W, got_b = ops.grouped_linear_wgrad_hprev(
        to_dev_bf16(hs), to_dev_bf16(dG), T
    )
    torch.testing.assert_close(got_W, want_W, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(got_b, want_b, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("F", [50, 7])  # 16B-vector path and scalar path
def test_window_gather_matches_torch(F):
    """K7 dedicated featurizer kernel vs the torch gather reference."""
    require_hip()
    G, N, B, T = 3, 500, 40, 144
    X = to_dev_bf16(_rand(G, N, F, seed=70))
    idx = torch.randint(0, N - T + 1, (G, B), device="cuda",
                        dtype=torch.int32)
    got = ops.window_gather(X, idx, T)
    rows = idx.long().unsqueeze(-1) + torch.arange(T, device="cuda")
    want = X.gather(
        1, rows.reshape(G, B * T, 1).expand(G, B * T, F)
    ).view(G, B, T, F)
    assert torch.equal(got, want)


def test_lstm_scan_row_tiles_bit_equal(monkeypatch):
    """Row-tile choice (16/32/64) is a pure scheduling knob: per-row
    math is identical, so outputs are bit-equal across tiles."""
    require_hip()
    G, B, T, H = 2, 96, 20, 42
    xW = to_dev_bf16(_rand(G, B, T, 4 * H, seed=80))
    Wh = to_dev_bf16(_rand(G, H, 4 * H, seed=81) * 0.3)
    outs = {}
    for rows in ("16", "32", "64"):
        monkeypatch.setenv("GORDO_LSTM_ROWS", rows)
        hs, cs, ga = ops.lstm_seq_fwd(xW, Wh)
        dSeq = to_dev_bf16(_rand(G, B, H, seed=82))
        dG = ops.lstm_seq_bwd(dSeq, ga, cs, Wh, True)
        outs[rows] = (hs, cs, ga, dG)
    monkeypatch.delenv("GORDO_LSTM_ROWS")
    for rows in ("32", "64"):
        for a, b in zip(outs["16"], outs[rows]):
            assert torch.equal(a, b), rows


@pytest.mark.parametrize("G,B,T,H,F", [
    (2, 48, 20, 48, 56),    # bench-like padded dims
    (1, 33, 12, 16, 8),     # minimal geometry, ragged rows
    (2, 64, 24, 64, 128),   # caps
])
def test_lstm_seq_v4_fused_vs_twostep(G, B, T, H, F):
    """The v4 fused-xW scan tracks the fp32 oracle (its PRIMARY
    correctness bar: v4 keeps the x-side contribution in the fp32
    accumulator, while the two-step path rounds xW to bf16 first, so
    v4-vs-twostep drift compounds over the recurrence and only a
    loose sanity bound applies between them). Inference mode
    (store_aux=False) is bit-identical to the training forward."""
    require_hip()
    x = to_dev_bf16(_rand(G, B, T, F, seed=90) * 0.5)
    Wx = to_dev_bf16(_rand(G, F, 4 * H, seed=91) * 0.2)
    Wh = to_dev_bf16(_rand(G, H, 4 * H, seed=92) * 0.2)
    b = _rand(G, 4 * H, seed=93).cuda() * 0.1

    hs4, cs4, ga4 = ops.lstm_seq_fwd_fused(x, Wx, Wh, b, store_aux=True)
    (hs_inf,) = ops.lstm_seq_fwd_fused(x, Wx, Wh, b, store_aux=False)
    assert torch.equal(hs_inf, hs4)

    # fp32 oracle through the same fused entry (primary)
    want_hs, want_cs, want_ga = ops.lstm_seq_fwd_fused(
        x.float().cpu(), Wx.float().cpu(), Wh.float().cpu(),
        b.float().cpu(), store_aux=True,
    )
    torch.testing.assert_close(
        hs4.float().cpu(), want_hs, rtol=6e-2, atol=3e-2
    )
    torch.testing.assert_close(
        cs4.cpu(), want_cs, rtol=6e-2, atol=5e-2
    )
    torch.testing.assert_close(
        ga4.float().cpu(), want_ga, rtol=6e-2, atol=3e-2
    )

    # two-step path sanity: v4 (fp32 x-accumulation) must track the
    # fp32 oracle at least as well as the two-step path (which rounds
    # xW to bf16 first) — an absolute bound between the two bf16 paths
    # is brittle at large K
    xW = ops.grouped_linear_fwd(
        x.reshape(G, B * T, F), Wx, b, "linear"
    ).view(G, B, T, 4 * H)
    hs1, _cs1, _ga1 = ops.lstm_seq_fwd(xW, Wh)
    err4 = (hs4.float().cpu() - want_hs).abs().mean().item()
    err1 = (hs1.float().cpu() - want_hs).abs().mean().item()
    assert err4 <= err1 * 1.5 + 0.01, (err4, err1)


def test_lstm_pack_v4_matches_twostep_end_to_end(monkeypatch):
    """LSTMPack training with the v4 fused scans tracks the two-step
    path (same pack, same seeds) within bf16 drift."""
    require_hip()
    from gordo_amd.engine.pack import LSTMPack
    from gordo_amd.engine.spec import LayerSpec, ModelSpec

    spec = ModelSpec(
        model_type="lstm", n_features=10, n_features_out=10,
        layers=[
            LayerSpec(kind="lstm", units=12, return_sequences=True),
            LayerSpec(kind="lstm", units=12, return_sequences=False),
            LayerSpec(kind="dense", units=10, activation="linear"),
        ],
        lookback_window=16,
    )
    X = torch.tensor(
        np.random.default_rng(5).random((1, 300, 10)).astype("float32")
    )
    losses = {}
    for arm, env in (("v4", "1"), ("twostep", "0")):
        monkeypatch.setenv("GORDO_LSTM_V4", env)
        p = LSTMPack(spec, G=1, device="cuda", seeds=[3])
        Xg = X.to("cuda", p.compute_dtype)
        h = p.fit(Xg, Xg.clone(), epochs=2, batch_size=64, shuffle=False)
        losses[arm] = [e[0] for e in h["loss"]]
    monkeypatch.delenv("GORDO_LSTM_V4")
    for a, b in zip(losses["v4"], losses["twostep"]):
        assert a == pytest.approx(b, rel=0.05, abs=1e-3)


@pytest.mark.parametrize("G,B,T,H,F,last_only", [
    (2, 48, 16, 48, 56, True),
    (1, 33, 10, 16, 16, False),
    (2, 40, 12, 64, 128, True),
])
def test_lstm_seq_bwd_v5_fused_dseq(G, B, T, H, F, last_only):
    """v5 fused-dSeq reverse scan: dG equals the plain bwd scan
    bit-for-bit (same math, only the carry MFMA widened) and dX equals
    the separate bwd-data GEMM within bf16 tolerance."""
    require_hip()
    x = to_dev_bf16(_rand(G, B, T, F, seed=95) * 0.5)
    Wx = to_dev_bf16(_rand(G, F, 4 * H, seed=96) * 0.2)
    Wh = to_dev_bf16(_rand(G, H, 4 * H, seed=97) * 0.2)
    b = _rand(G, 4 * H, seed=98).cuda() * 0.1
    hs, cs, ga = ops.lstm_seq_fwd_fused(x, Wx, Wh, b, store_aux=True)
    dSeq = to_dev_bf16(
        _rand(G, B, H, seed=99) if last_only
        else _rand(G, B, T, H, seed=99)
    )
    dG_ref = ops.lstm_seq_bwd(dSeq, ga, cs, Wh, last_only)
    dX_ref = ops.grouped_linear_bwd_data(
        dG_ref.view(G, B * T, 4 * H), Wx
    ).view(G, B, T, F)
    dG5, dX5 = ops.lstm_seq_bwd_fused(dSeq, ga, cs, Wh, Wx, last_only)
    torch.testing.assert_close(
        dG5.float(), dG_ref.float(), rtol=3e-2, atol=1e-2
    )
    torch.testing.assert_close(
        dX5.float(), dX_ref.float(), rtol=5e-2, atol=2e-2
    )


@pytest.mark.parametrize("G,B,T,F,H,N4", [(3, 16, 12, 50, 42, 168),
                                          (1, 9, 7, 8, 16, 64)])
def test_grouped_wgrad_xh_matches_separate(G, B, T, F, H, N4):
    """The combined {dWx, dWh, db} pass equals the two separate wgrad
    calls (one dZ staging instead of two)."""
    require_hip()
    seq = to_dev_bf16(_rand(G, B * T, F, seed=64))
    hs = to_dev_bf16(_rand(G, B, T, H, seed=65))
    dG = to_dev_bf16(_rand(G, B * T, N4, seed=66))
    want_Wx, want_b = ops.grouped_linear_wgrad(seq, dG)
    want_Wh, _ = ops.grouped_linear_wgrad_hprev(hs, dG, T)
    got_Wx, got_Wh, got_b = ops.grouped_wgrad_xh(seq, hs, dG, T)
    torch.testing.assert_close(got_Wx, want_Wx, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(got_Wh, want_Wh, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(got_b, want_b, rtol=1e-5, atol=1e-5)


@pytest.mark.gpu
@pytest.mark.parametrize("rows", [16, 32, 64])
def test_lstm_seq_v4_fused_deterministic(rows, monkeypatch):
    """Repeated v4 launches are bitwise identical at every row tile.

    Regression for an LDS init race: the hxS zero-init loop covered the
    x section that the x_0 staging loop writes concurrently (different
    threads own the same address in the two loops), corrupting t=0
    gates under workgroup co-residency — nondeterministic across
    launches (up to 12/30 differing before the fix)."""
    require_hip()
    monkeypatch.setenv("GORDO_LSTM_ROWS", str(rows))
    G, B, T, H, F = 2, 64, 24, 64, 128
    x = to_dev_bf16(_rand(G, B, T, F, seed=90) * 0.5)
    Wx = to_dev_bf16(_rand(G, F, 4 * H, seed=91) * 0.2)
    Wh = to_dev_bf16(_rand(G, H, 4 * H, seed=92) * 0.2)
    b = _rand(G, 4 * H, seed=93).cuda() * 0.1
    ref = ops.lstm_seq_fwd_fused(x, Wx, Wh, b, store_aux=True)
    for _ in range(20):
        out = ops.lstm_seq_fwd_fused(x, Wx, Wh, b, store_aux=True)
        for a, c in zip(ref, out):
            assert torch.equal(a, c)
        (hi,) = ops.lstm_seq_fwd_fused(x, Wx, Wh, b, store_aux=False)
        assert torch.equal(hi, ref[0])


@pytest.mark.gpu
@pytest.mark.parametrize(
    "name,env,G,B,H",
    [
        ("v1", {"GORDO_LSTM_V1": "1"}, 2, 64, 48),
        ("v3", {}, 2, 64, 42),                  # default pipelined path
        ("v2", {}, 40, 512, 48),                # H%16==0 + filled grid
        ("big", {}, 2, 64, 128),                # streamed-Wh H>64 path
    ],
)
def test_lstm_seq_scan_deterministic(name, env, G, B, H, monkeypatch):
    """Every scan dispatch path (v1/v2/v3/big-H) is bitwise
    deterministic across repeated forward AND backward launches —
    companion to the v4 determinism regression."""
    require_hip()
    for k, v in env.items():
        monkeypatch.setenv(k, v)
    T, H4 = 16, 4 * H
    g = torch.Generator().manual_seed(7)
    xW = (torch.randn(G, B, T, H4, generator=g) * 0.3).to(
        "cuda", torch.bfloat16
    )
    Wh = (torch.randn(G, H, H4, generator=g) * 0.2).to(
        "cuda", torch.bfloat16
    )
    hs, cs, ga = ops.lstm_seq_fwd(xW, Wh)
    dSeq = (torch.randn(G, B, T, H, generator=g) * 0.1).to(
        "cuda", torch.bfloat16
    )
    dref = ops.lstm_seq_bwd(dSeq, ga, cs, Wh, last_only=False)
    for _ in range(10):
        out = ops.lstm_seq_fwd(xW, Wh)
        for a, c in zip((hs, cs, ga), out):
            assert torch.equal(a, c), name
        d = ops.lstm_seq_bwd(dSeq, ga, cs, Wh, last_only=False)
        assert torch.equal(d, dref), name
