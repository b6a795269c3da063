import numpy as np
import pandas as pd
import pytest
from sklearn.pipeline import Pipeline
from sklearn.preprocessing import MinMaxScaler

from gordo_amd.machine.model.anomaly.diff import (
    DiffBasedAnomalyDetector,
    DiffBasedKFCVAnomalyDetector,
)
from gordo_amd.machine.model.models import KerasAutoEncoder


@pytest.fixture(scope="module")
def Xy():
    rng = np.random.default_rng(42)
    t = np.linspace(0, 30, 400)
    X = pd.DataFrame(
        {f"tag-{i}": np.sin(t + i) + rng.normal(0, 0.05, len(t)) for i in range(4)}
    )
    return X, X.copy()


def _detector(**kwargs):
    return DiffBasedAnomalyDetector(
        base_estimator=Pipeline(
            [
                ("mms", MinMaxScaler()),
                ("ae", KerasAutoEncoder(kind="feedforward_hourglass", epochs=1)),
            ]
        ),
        **kwargs,
    )


def test_fit_and_anomaly_columns(Xy):
    X, y = Xy
    det = _detector(require_thresholds=False)
    det.fit(X, y)
    frame = det.anomaly(X, y, frequency=pd.Timedelta(minutes=10))
    tops = {c[0] for c in frame.columns}
    assert {
        "start", "end", "model-input", "model-output",
        "tag-anomaly-scaled", "total-anomaly-scaled",
        "tag-anomaly-unscaled", "total-anomaly-unscaled",
    } <= tops
    # without cross_validate, no confidence columns
    assert "anomaly-confidence" not in tops
    assert "total-anomaly-confidence" not in tops

    # verify scaled tag anomaly against hand-computed pandas
    pred = det.base_estimator.predict(X.values)
    expected = np.abs(det.scaler.transform(pred) - det.scaler.transform(y))
    np.testing.assert_allclose(
        frame["tag-anomaly-scaled"].values, expected, rtol=1e-5, atol=1e-6
    )
    expected_total = (expected ** 2).mean(axis=1)
    np.testing.assert_allclose(
        frame["total-anomaly-scaled"].values, expected_total, rtol=1e-5,
        atol=1e-6,
    )


def test_require_thresholds_raises(Xy):
    X, y = Xy
    det = _detector(require_thresholds=True)
    det.fit(X, y)
    with pytest.raises(AttributeError):
        det.anomaly(X, y)


def test_cross_validate_sets_thresholds(Xy):
    X, y = Xy
    det = _detector(require_thresholds=True)
    det.cross_validate(X=X, y=y)
    det.fit(X, y)
    assert hasattr(det, "aggregate_threshold_")
    assert hasattr(det, "feature_thresholds_")
    assert len(det.feature_thresholds_) == X.shape[1]
    assert len(det.aggregate_thresholds_per_fold_) == 3  # TimeSeriesSplit(3)
    frame = det.anomaly(X, y, frequency=pd.Timedelta(minutes=10))
    tops = {c[0] for c in frame.columns}
    assert "anomaly-confidence" in tops
    assert "total-anomaly-confidence" in tops
    np.testing.assert_allclose(
        frame["total-anomaly-confidence"].values,
        frame["total-anomaly-scaled"].values / det.aggregate_threshold_,
        rtol=1e-6,
    )


def test_smoothing_columns(Xy):
    X, y = Xy
    det = _detector(require_thresholds=False, window=10)
    assert det.smoothing_method == "smm"  # default when window given
    det.fit(X, y)
    frame = det.anomaly(X, y)
    tops = {c[0] for c in frame.columns}
    assert "smooth-tag-anomaly-scaled" in tops
    assert "smooth-total-anomaly-scaled" in tops
    # smm == rolling median of the raw series
    raw = frame["total-anomaly-scaled"]
    np.testing.assert_allclose(
        frame["smooth-total-anomaly-scaled"].values,
        raw.rolling(10).median().values,
        rtol=1e-6, equal_nan=True,
    )


@pytest.mark.parametrize("method,window", [("sma", 5), ("ewma", 5)])
def test_smoothing_methods(Xy, method, window):
    X, y = Xy
    det = _detector(require_thresholds=False, window=window,
                    smoothing_method=method)
    det.fit(X, y)
    frame = det.anomaly(X, y)
    raw = frame["total-anomaly-scaled"]
    if method == "sma":
        expected = raw.rolling(window).mean()
    else:
        expected = raw.ewm(span=window).mean()
    np.testing.assert_allclose(
        frame["smooth-total-anomaly-scaled"].values, expected.values,
        rtol=1e-6, equal_nan=True,
    )


def test_kfcv_detector(Xy):
    X, y = Xy
    det = DiffBasedKFCVAnomalyDetector(
        base_estimator=Pipeline(
            [
                ("mms", MinMaxScaler()),
                ("ae", KerasAutoEncoder(kind="feedforward_hourglass", epochs=1)),
            ]
        ),
        window=20,
        threshold_percentile=0.99,
    )
    det.cross_validate(X=X, y=y)
    det.fit(X, y)
    assert np.isfinite(det.aggregate_threshold_)
    assert len(det.feature_thresholds_) == X.shape[1]
    frame = det.anomaly(X, y)
    assert "total-anomaly-confidence" in {c[0] for c in frame.columns}


def test_getattr_passthrough():
    det = _detector()
    # attribute of the base estimator reachable through the detector
    assert det.steps[1][0] == "ae"


def test_get_params_roundtrip():
    det = _detector(require_thresholds=False, window=12)
    params = det.get_params()
    assert params["window"] == 12
    assert params["smoothing_method"] == "smm"
    det2 = DiffBasedAnomalyDetector(**params)
    assert det2.window == 12


@pytest.mark.parametrize("mode", ["tscv", "tscv_win", "kfcv"])
def test_diff_detector_metadata_lifecycle(mode):
    """get_metadata keys before and after cross_validate (reference
    test_anomaly_detectors.py::test_diff_detector_get_metadata)."""
    import pandas as pd
    from sklearn.linear_model import LinearRegression
    from sklearn.multioutput import MultiOutputRegressor

    from gordo_amd.machine.model.anomaly.diff import (
        DiffBasedAnomalyDetector,
        DiffBasedKFCVAnomalyDetector,
    )

    X = pd.DataFrame(np.random.RandomState(0).random((200, 5)))
    y = pd.DataFrame(np.random.RandomState(1).random((200, 2)))
    base = MultiOutputRegressor(LinearRegression())
    if mode == "tscv":
        model = DiffBasedAnomalyDetector(base_estimator=base)
    elif mode == "tscv_win":
        model = DiffBasedAnomalyDetector(base_estimator=base, window=144)
    else:
        model = DiffBasedKFCVAnomalyDetector(base_estimator=base)

    md = model.get_metadata()
    assert "base_estimator" in md and "scaler" in md and "shuffle" in md
    for key in (
        "feature-thresholds", "aggregate-threshold",
        "feature-thresholds-per-fold", "aggregate-thresholds-per-fold",
    ):
        assert key not in md

    # thresholds appear only after the detector's own cross_validate
    assert not hasattr(model, "feature_thresholds_")
    model.fit(X, y)
    assert not hasattr(model, "feature_thresholds_")
    model.cross_validate(X=X, y=y)
    md = model.get_metadata()

    assert isinstance(md["feature-thresholds"], list)
    assert len(md["feature-thresholds"]) == 2
    assert "aggregate-threshold" in md
    assert isinstance(model.feature_thresholds_, pd.Series)
    assert model.feature_thresholds_.notna().all()

    if mode != "kfcv":
        assert "feature-thresholds-per-fold" in md
        assert "aggregate-thresholds-per-fold" in md
        assert isinstance(model.feature_thresholds_per_fold_, pd.DataFrame)
    if mode != "tscv":
        assert "window" in md
        assert "smoothing-method" in md
    if mode == "tscv_win":
        for key in (
            "smooth-feature-thresholds",
            "smooth-aggregate-threshold",
            "smooth-feature-thresholds-per-fold",
            "smooth-aggregate-thresholds-per-fold",
        ):
            assert key in md, key
    if mode == "kfcv":
        assert "threshold-percentile" in md


def test_anomaly_frequency_parameter(Xy):
    """anomaly(..., frequency=) sets the `end` column spacing (the
    serving path passes the dataset resolution)."""
    import pandas as pd

    X, y = Xy
    X = X.copy()
    X.index = pd.date_range("2020-01-01", periods=len(X), freq="10min",
                            tz="UTC")
    y = X.copy()
    det = _detector(require_thresholds=False)
    det.fit(X, y)
    out = det.anomaly(X, y, frequency=pd.Timedelta("10min"))
    start = pd.to_datetime(out["start"].iloc[0], utc=True)
    end = pd.to_datetime(out["end"].iloc[0], utc=True)
    assert (end - start) == pd.Timedelta("10min")
