"""Serving micro-batcher: concurrency equivalence + self-detection."""
import threading

import numpy as np

from gordo_amd.server.batcher import MicroBatcher


class CountingModel:
    def __init__(self):
        self.calls = 0
        self.lock = threading.Lock()

    def predict(self, X):
        with self.lock:
            self.calls += 1
        return np.asarray(X) * 2.0


class WindowedModel(CountingModel):
    """Output shorter than input (LSTM-like) — must NOT be batched."""

    def predict(self, X):
        with self.lock:
            self.calls += 1
        return np.asarray(X)[3:] * 2.0


def test_batcher_results_identical_under_concurrency():
    model = CountingModel()
    b = MicroBatcher(model.predict, window_ms=20.0)
    b.predict(np.zeros((2, 4)))  # probe establishes alignment

    rng = np.random.default_rng(0)
    inputs = [rng.random((10 + i, 4)) for i in range(12)]
    outputs = [None] * len(inputs)

    def run(i):
        outputs[i] = b.predict(inputs[i])

    threads = [threading.Thread(target=run, args=(i,))
               for i in range(len(inputs))]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    for i, (x, y) in enumerate(zip(inputs, outputs)):
        np.testing.assert_allclose(y, x * 2.0, err_msg=str(i))
    # coalescing happened: far fewer predict calls than requests
    assert model.calls < 1 + len(inputs)


def test_batcher_bypasses_windowed_models():
    model = WindowedModel()
    b = MicroBatcher(model.predict, window_ms=1.0)
    X = np.random.default_rng(1).random((10, 4))
    out = b.predict(X)  # probe detects misalignment
    assert len(out) == 7
    out2 = b.predict(X)  # subsequent calls bypass coalescing
    np.testing.assert_allclose(out2, X[3:] * 2.0)
    assert b._aligned is False


def test_batcher_row_budget_and_shape_mismatch():
    model = CountingModel()
    b = MicroBatcher(model.predict, window_ms=10.0, max_rows=16)
    b.predict(np.zeros((2, 4)))

    wide = np.random.default_rng(2).random((4, 6))  # different n_features
    big = np.random.default_rng(3).random((15, 4))
    results = {}

    def run(name, X):
        results[name] = b.predict(X)

    threads = [
        threading.Thread(target=run, args=("a", big)),
        threading.Thread(target=run, args=("b", big)),   # exceeds budget
        threading.Thread(target=run, args=("c", wide)),  # shape mismatch
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    np.testing.assert_allclose(results["a"], big * 2.0)
    np.testing.assert_allclose(results["b"], big * 2.0)
    np.testing.assert_allclose(results["c"], wide * 2.0)


def test_batched_endpoint_equivalence(api_client, base_route, monkeypatch):
    """/prediction responses identical with the batcher enabled."""
    import pandas as pd

    from gordo_amd.server import utils as server_utils

    rng = np.random.default_rng(4)
    sensors = None
    # reuse the suite's payload helper shape
    resp_plain = None
    monkeypatch.delenv("GORDO_SERVE_BATCH", raising=False)
    X = None
    # fetch tags from metadata to build a valid payload
    md = api_client.get(f"{base_route}/metadata").json["metadata"]
    tags = [t["name"] for t in md["dataset"]["tag_list"]]
    X = pd.DataFrame(rng.random((20, len(tags))), columns=tags,
                     index=pd.date_range("2019-01-01", periods=20,
                                         freq="10min", tz="UTC"))
    payload = {"X": server_utils.dataframe_to_dict(X)}
    resp_plain = api_client.post(f"{base_route}/prediction", json=payload)
    assert resp_plain.status_code == 200

    monkeypatch.setenv("GORDO_SERVE_BATCH", "1")
    resp_batched = api_client.post(f"{base_route}/prediction", json=payload)
    assert resp_batched.status_code == 200
    assert resp_batched.json["data"] == resp_plain.json["data"]
