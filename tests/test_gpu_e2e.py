"""End-to-end GPU tests: packed training through the HIP path, estimator
fit on cuda, GPU-vs-CPU build equivalence at bf16 tolerance."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def require_hip():
    from gordo_amd import ops

    assert ops.hip_available(), "HIP extension not built"


def test_dense_pack_gpu_matches_cpu():
    require_hip()
    from gordo_amd.engine.pack import DensePack
    from gordo_amd.engine.spec import LayerSpec, ModelSpec

    spec = ModelSpec(
        model_type="feedforward", n_features=50, n_features_out=50,
        layers=[
            LayerSpec(kind="dense", units=38, activation="tanh"),
            LayerSpec(kind="dense", units=25, activation="tanh",
                      l1_activity=1e-4),
            LayerSpec(kind="dense", units=38, activation="tanh"),
            LayerSpec(kind="dense", units=50, activation="linear"),
        ],
    )
    rng = np.random.default_rng(0)
    X = rng.random((2, 512, 50)).astype("float32")
    Xt = torch.from_numpy(X)

    cpu = DensePack(spec, G=2, device="cpu", seeds=[1, 2])
    hist_cpu = cpu.fit(Xt, Xt.clone(), epochs=3, batch_size=128)

    gpu = DensePack(spec, G=2, device="cuda", seeds=[1, 2])
    Xg = Xt.to("cuda", gpu.compute_dtype)
    hist_gpu = gpu.fit(Xg, Xg.clone(), epochs=3, batch_size=128)

    # losses track within bf16 drift
    for e in range(3):
        for g in range(2):
            assert hist_gpu["loss"][e][g] == pytest.approx(
                hist_cpu["loss"][e][g], rel=0.08, abs=2e-3
            )
    # predictions close
    out_cpu = cpu.predict(Xt[:, :64]).float().numpy()
    out_gpu = gpu.predict(Xt[:, :64]).float().cpu().numpy()
    assert np.abs(out_cpu - out_gpu).mean() < 0.03


def test_lstm_pack_gpu_matches_cpu():
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
    rng = np.random.default_rng(1)
    X = rng.random((1, 400, 10)).astype("float32")
    Xt = torch.from_numpy(X)

    cpu = LSTMPack(spec, G=1, device="cpu", seeds=[3])
    hist_cpu = cpu.fit(Xt, Xt.clone(), epochs=2, batch_size=64)
    gpu = LSTMPack(spec, G=1, device="cuda", seeds=[3])
    Xg = Xt.to("cuda", gpu.compute_dtype)
    hist_gpu = gpu.fit(Xg, Xg.clone(), epochs=2, batch_size=64)
    for e in range(2):
        assert hist_gpu["loss"][e][0] == pytest.approx(
            hist_cpu["loss"][e][0], rel=0.1, abs=3e-3
        )
    out_cpu = cpu.predict(Xt).float().numpy()
    out_gpu = gpu.predict(Xt).float().cpu().numpy()
    assert out_gpu.shape == out_cpu.shape
    assert np.abs(out_cpu - out_gpu).mean() < 0.05


def test_lstm_pack_default_dims_gpu_matches_cpu():
    """The reference's DEFAULT LSTM dims (256, 128, 64) — reference
    lstm_autoencoder.py:112 — exercise the big-H (H > 64) fused scan
    kernels end to end through LSTMPack."""
    require_hip()
    from gordo_amd.engine.pack import LSTMPack
    from gordo_amd.engine.spec import LayerSpec, ModelSpec

    spec = ModelSpec(
        model_type="lstm", n_features=10, n_features_out=10,
        layers=[
            LayerSpec(kind="lstm", units=256, return_sequences=True),
            LayerSpec(kind="lstm", units=128, return_sequences=True),
            LayerSpec(kind="lstm", units=64, return_sequences=False),
            LayerSpec(kind="dense", units=10, activation="linear"),
        ],
        lookback_window=12,
    )
    rng = np.random.default_rng(7)
    X = rng.random((1, 160, 10)).astype("float32")
    Xt = torch.from_numpy(X)

    gpu = LSTMPack(spec, G=1, device="cuda", seeds=[5])
    # grid-fill heuristic: big-H stacks go fused only when the scan
    # grid fills the chip (G*ceil(B/32) >= 256)
    assert gpu._use_fused(8192), "full grid must take the fused path"
    assert not gpu._use_fused(32), "underfilled big-H grid must fall back"
    cpu = LSTMPack(spec, G=1, device="cpu", seeds=[5])
    hist_cpu = cpu.fit(Xt, Xt.clone(), epochs=2, batch_size=64)
    Xg = Xt.to("cuda", gpu.compute_dtype)
    hist_gpu = gpu.fit(Xg, Xg.clone(), epochs=2, batch_size=64)
    for e in range(2):
        assert hist_gpu["loss"][e][0] == pytest.approx(
            hist_cpu["loss"][e][0], rel=0.15, abs=5e-3
        )
    out_cpu = cpu.predict(Xt).float().numpy()
    out_gpu = gpu.predict(Xt).float().cpu().numpy()
    assert out_gpu.shape == out_cpu.shape
    assert np.abs(out_cpu - out_gpu).mean() < 0.05


def test_predict_captured_matches_eager(monkeypatch):
    """hipGraph-captured serving forward (GORDO_SERVE_HIPGRAPH=1)
    replays bit-identically to the eager forward, across shapes and
    repeated replays."""
    require_hip()
    from gordo_amd.machine.model import KerasAutoEncoder

    X = np.random.default_rng(11).random((300, 20)).astype("float32")
    model = KerasAutoEncoder(kind="feedforward_hourglass", epochs=2,
                             batch_size=64)
    model.fit(X)
    eager = model.predict(X[:100])
    eager2 = model.predict(X[:50])
    monkeypatch.setenv("GORDO_SERVE_HIPGRAPH", "1")
    cap_first = model.predict(X[:100])     # capture
    cap_replay = model.predict(X[100:200])  # replay, new data
    cap_other = model.predict(X[:50])      # second shape bucket
    assert np.array_equal(cap_first, eager)
    assert np.array_equal(cap_other, eager2)
    monkeypatch.delenv("GORDO_SERVE_HIPGRAPH")
    assert np.array_equal(cap_replay, model.predict(X[100:200]))
    pack = model._pack
    assert len(pack._pred_graph_cache) == 2
    pack.release_graphs()


def test_estimator_fit_on_gpu_uses_hip():
    require_hip()
    from gordo_amd.machine.model import KerasAutoEncoder

    X = np.random.default_rng(2).random((300, 20)).astype("float32")
    model = KerasAutoEncoder(kind="feedforward_hourglass", epochs=5,
                             batch_size=64)
    model.fit(X)
    assert model._pack.device.type == "cuda"
    out = model.predict(X)
    assert out.shape == X.shape
    assert np.isfinite(out).all()
    # training reduces loss on structured data
    t = np.linspace(0, 20, 300)
    signal = np.stack([np.sin(t + p) for p in np.linspace(0, 1, 20)], axis=1)
    m2 = KerasAutoEncoder(kind="feedforward_hourglass", epochs=15,
                          batch_size=64)
    m2.fit(signal.astype("float32"))
    losses = m2.history["loss"]
    assert losses[-1] < losses[0] * 0.5


def test_packed_fleet_builder_on_gpu(tmp_path):
    require_hip()
    from gordo_amd.parallel import PackedFleetBuilder
    from gordo_amd.workflow import NormalizedConfig

    machines = [
        {
            "name": f"gpu-m-{i}",
            "dataset": {
                "type": "SineWaveDataset",
                "tag_list": [f"t-{j}" for j in range(20)],
                "train_start_date": "2019-01-01T00:00:00+00:00",
                "train_end_date": "2019-01-08T00:00:00+00:00",
            },
            "model": {
                "gordo_amd.machine.model.anomaly.diff.DiffBasedAnomalyDetector": {
                    "require_thresholds": True,
                    "base_estimator": {
                        "sklearn.pipeline.Pipeline": {
                            "steps": [
                                "sklearn.preprocessing.MinMaxScaler",
                                {
                                    "gordo_amd.machine.model.models.KerasAutoEncoder": {
                                        "kind": "feedforward_hourglass",
                                        "epochs": 2,
                                    }
                                },
                            ]
                        }
                    },
                }
            },
        }
        for i in range(8)
    ]
    norm = NormalizedConfig({"machines": machines}, project_name="gpu-fleet")
    fb = PackedFleetBuilder(
        norm.machines, output_dir=str(tmp_path), device="cuda"
    )
    results = dict(fb.build_all())
    assert all(not isinstance(v, BaseException) for v in results.values()), {
        k: repr(v) for k, v in results.items() if isinstance(v, BaseException)
    }
    # serve one model back (CPU predict path from pickle)
    import pandas as pd

    from gordo_amd import serializer

    model = serializer.load(str(tmp_path / "gpu-m-0"))
    X = pd.DataFrame(
        np.random.rand(100, 20), columns=[f"t-{j}" for j in range(20)]
    )
    frame = model.anomaly(X, X)
    assert "total-anomaly-confidence" in {c[0] for c in frame.columns}


def test_early_stopping_and_validation_split_on_gpu():
    """The fit-loop additions (EarlyStopping patience, validation_split
    holdout + val_loss) behave on the HIP engine exactly as on CPU."""
    import numpy as np

    from gordo_amd.machine.model.models import KerasAutoEncoder

    X = np.random.RandomState(0).random((256, 12))
    model = KerasAutoEncoder(
        kind="feedforward_hourglass", epochs=40, batch_size=64,
        validation_split=0.25,
        callbacks=[{"tensorflow.keras.callbacks.EarlyStopping": {
            "monitor": "val_loss", "patience": 2, "min_delta": 1e9}}],
    )
    model.fit(X)
    hist = model.get_metadata()["history"]
    assert len(hist["loss"]) == 3  # stopped by patience, not epochs
    assert len(hist["val_loss"]) == 3
    assert all(np.isfinite(v) for v in hist["val_loss"])
    out = model.predict(X)
    assert out.shape == X.shape


def test_serving_response_encoder_on_gpu_box():
    """The C++ response encoder .so built on the CPU host loads and is
    byte-identical on the GPU box too (it travels with the snapshot)."""
    import json

    import numpy as np
    import pandas as pd

    from gordo_amd.server import _gordo_fastjson as fj
    from gordo_amd.server import utils as su

    df = pd.DataFrame(
        np.random.default_rng(0).random((50, 8)),
        columns=pd.MultiIndex.from_product(
            (("model-input", "model-output"), [f"t{i}" for i in range(4)])
        ),
        index=pd.date_range("2020-01-01", periods=50, freq="10min"),
    )
    fast = fj.encode_frame(
        df.index.astype(str).tolist(),
        [c[0] for c in df.columns],
        [c[1] for c in df.columns],
        df.values,
    )
    assert fast == json.dumps(su.dataframe_to_dict(df)).encode()
