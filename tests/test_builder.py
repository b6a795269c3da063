import dateutil.parser
import numpy as np
import pytest

from gordo_amd import serializer
from gordo_amd.builder import ModelBuilder, local_build
from gordo_amd.machine import Machine


def get_random_data():
    return {
        "type": "RandomDataset",
        "train_start_date": dateutil.parser.isoparse("2017-12-25 06:00:00Z"),
        "train_end_date": dateutil.parser.isoparse("2017-12-27 06:00:00Z"),
        "tag_list": ["Tag 1", "Tag 2"],
        "target_tag_list": ["Tag 1", "Tag 2"],
    }


KERAS_MODEL = {
    "gordo.machine.model.anomaly.diff.DiffBasedAnomalyDetector": {
        "require_thresholds": False,
        "base_estimator": {
            "sklearn.pipeline.Pipeline": {
                "steps": [
                    "sklearn.preprocessing.MinMaxScaler",
                    {
                        "gordo.machine.model.models.KerasAutoEncoder": {
                            "kind": "feedforward_hourglass",
                            "epochs": 1,
                        }
                    },
                ]
            }
        },
    }
}

SKLEARN_MODEL = {
    "sklearn.pipeline.Pipeline": {
        "steps": [
            "sklearn.preprocessing.MinMaxScaler",
            {"sklearn.decomposition.PCA": {"n_components": 2}},
        ]
    }
}


def make_machine(model=None, name="test-model", evaluation=None):
    cfg = dict(name=name, model=model or KERAS_MODEL, dataset=get_random_data())
    if evaluation:
        cfg["evaluation"] = evaluation
    return Machine.from_config(cfg, project_name="test-proj")


def machine_check(machine, check_history=False):
    """Metadata completeness (mirrors reference
    tests/gordo/builder/test_builder.py machine_check)."""
    md = machine.metadata.build_metadata.model
    assert isinstance(md.model_offset, int)
    scores = md.cross_validation.scores
    if scores:
        tag_list = [t.name.replace(" ", "-") for t in machine.dataset.tag_list]
        scores_list = [
            "r2-score",
            "explained-variance-score",
            "mean-squared-error",
            "mean-absolute-error",
        ]
        all_scores = [
            f"{s}-{t}" for s in scores_list for t in tag_list
        ] + scores_list
        assert all(s in scores for s in all_scores)
    if check_history:
        assert "history" in md.model_meta
        assert all(
            k in md.model_meta["history"] for k in ("params", "loss", "accuracy")
        )


def test_sklearn_model_build(tmp_path):
    machine = make_machine(model=SKLEARN_MODEL)
    model, machine_out = ModelBuilder(machine).build(output_dir=str(tmp_path))
    machine_check(machine_out)
    assert (tmp_path / "model.pkl").is_file()
    assert (tmp_path / "metadata.json").is_file()
    loaded = serializer.load(str(tmp_path))
    assert hasattr(loaded, "transform")


def test_keras_model_build(tmp_path):
    machine = make_machine()
    model, machine_out = ModelBuilder(machine).build(output_dir=str(tmp_path))
    machine_check(machine_out, check_history=True)
    md = machine_out.metadata.build_metadata
    assert md.model.model_training_duration_sec > 0
    assert md.dataset.query_duration_sec > 0
    assert md.model.model_builder_version
    # saved metadata round-trips to a Machine
    meta = serializer.load_metadata(str(tmp_path))
    m2 = Machine.from_dict(meta)
    assert m2.name == machine.name


def test_cv_mode_cross_val_only():
    machine = make_machine(evaluation={"cv_mode": "cross_val_only"})
    model, machine_out = ModelBuilder(machine).build()
    # scores present but model not trained on full data
    assert machine_out.metadata.build_metadata.model.cross_validation.scores
    assert (
        machine_out.metadata.build_metadata.model.model_training_duration_sec
        is None
    )


def test_cache_hit_and_miss(tmp_path):
    reg = tmp_path / "reg"
    out1 = tmp_path / "out1"
    machine = make_machine(model=SKLEARN_MODEL)
    b1 = ModelBuilder(machine)
    b1.build(output_dir=str(out1), model_register_dir=str(reg))
    assert b1.cached_model_path == str(out1)

    # same machine → cache hit (no new output dir writes needed)
    b2 = ModelBuilder(make_machine(model=SKLEARN_MODEL))
    out2 = tmp_path / "out2"
    b2.build(output_dir=str(out2), model_register_dir=str(reg))
    assert b2.cached_model_path == str(out1)

    # different name → different key → miss
    b3 = ModelBuilder(make_machine(model=SKLEARN_MODEL, name="other-model"))
    out3 = tmp_path / "out3"
    b3.build(output_dir=str(out3), model_register_dir=str(reg))
    assert b3.cached_model_path == str(out3)

    # replace_cache forces rebuild
    b4 = ModelBuilder(make_machine(model=SKLEARN_MODEL))
    out4 = tmp_path / "out4"
    b4.build(output_dir=str(out4), model_register_dir=str(reg),
             replace_cache=True)
    assert b4.cached_model_path == str(out4)


def test_cache_key_stability():
    m1 = make_machine(model=SKLEARN_MODEL)
    m2 = make_machine(model=SKLEARN_MODEL)
    assert ModelBuilder(m1).cache_key == ModelBuilder(m2).cache_key
    assert len(ModelBuilder(m1).cache_key) == 128
    m3 = make_machine(model=SKLEARN_MODEL, name="other-model")
    assert ModelBuilder(m3).cache_key != ModelBuilder(m1).cache_key


def test_seed_determinism():
    machine1 = make_machine(evaluation={"cv_mode": "full_build", "seed": 42})
    machine2 = make_machine(evaluation={"cv_mode": "full_build", "seed": 42})
    model1, _ = ModelBuilder(machine1).build()
    model2, _ = ModelBuilder(machine2).build()
    X = np.random.RandomState(0).rand(30, 2)
    # GPU training is deterministic up to fp32-atomic accumulation
    # order in the split-M weight-grad kernel; CPU is bitwise
    import torch

    tol = dict(rtol=5e-2, atol=5e-3) if torch.cuda.is_available() else dict(
        rtol=1e-5, atol=1e-6
    )
    np.testing.assert_allclose(model1.predict(X), model2.predict(X), **tol)


def test_metrics_from_list():
    metrics = ModelBuilder.metrics_from_list(
        ["sklearn.metrics.r2_score", "mean_absolute_error"]
    )
    assert [m.__name__ for m in metrics] == ["r2_score", "mean_absolute_error"]


def test_determine_offset():
    machine = make_machine(
        model={
            "gordo.machine.model.models.KerasLSTMAutoEncoder": {
                "kind": "lstm_hourglass",
                "lookback_window": 5,
                "epochs": 1,
            }
        },
        evaluation={"cv_mode": "full_build"},
    )
    _, machine_out = ModelBuilder(machine).build()
    assert machine_out.metadata.build_metadata.model.model_offset == 4


def test_local_build(config_str):
    results = list(local_build(config_str))
    assert len(results) == 2
    for model, machine in results:
        machine_check(machine)
        assert machine.project_name == "local-build"


from sklearn.base import BaseEstimator as _SkBase


class _ScalingRegressor(_SkBase):
    """Predicts X * multiplier — used to probe per-feature error scaling."""

    def __init__(self, multiplier):
        self.multiplier = multiplier

    def fit(self, X, y):
        return self

    def predict(self, X):
        return np.asarray(X) * self.multiplier


@pytest.mark.parametrize(
    "scaler", [None, "sklearn.preprocessing.MinMaxScaler"]
)
def test_build_metrics_dict_scaler(scaler):
    """With a scoring scaler, a 20% error on a large-scale feature and on
    a small-scale feature weigh the same (reference
    tests/gordo/builder/test_builder.py::test_get_metrics_dict_scaler)."""
    import pandas as pd
    import sklearn.metrics

    y = pd.DataFrame(
        np.array([[1, 1], [2, 2], [3, 3], [4, 4], [5, 5]]) * [1, 100],
        columns=["Tag 1", "Tag 2"],
    )
    metrics_dict = ModelBuilder.build_metrics_dict(
        [sklearn.metrics.mean_squared_error], y, scaler=scaler
    )
    metric_func = metrics_dict["mean-squared-error"]
    mse_feat1_wrong = metric_func(_ScalingRegressor(np.array([0.8, 1.0])), y, y)
    mse_feat2_wrong = metric_func(_ScalingRegressor(np.array([1.0, 0.8])), y, y)
    if scaler:
        assert np.isclose(mse_feat1_wrong, mse_feat2_wrong)
    else:
        assert not np.isclose(mse_feat1_wrong, mse_feat2_wrong)


def test_determine_offset_helper():
    from sklearn.linear_model import LinearRegression
    from sklearn.multioutput import MultiOutputRegressor

    from gordo_amd.machine.model import models

    X = np.random.RandomState(0).random((100, 10))
    y = np.random.RandomState(1).random((100, 10))
    cases = [
        (
            models.KerasLSTMAutoEncoder(
                kind="lstm_hourglass", lookback_window=10, epochs=1
            ),
            9,
        ),
        (
            models.KerasLSTMForecast(
                kind="lstm_symmetric", lookback_window=13, epochs=1
            ),
            13,
        ),
        (models.KerasAutoEncoder(kind="feedforward_hourglass", epochs=1), 0),
        (MultiOutputRegressor(LinearRegression()), 0),
    ]
    for model, expected in cases:
        model.fit(X, y)
        assert ModelBuilder._determine_offset(model, X) == expected, model


def test_output_dir_nested(tmp_path):
    """Builder creates missing intermediate directories."""
    out = tmp_path / "some" / "sub" / "directories"
    machine = make_machine(model=SKLEARN_MODEL)
    ModelBuilder(machine).build(output_dir=str(out))
    assert (out / "model.pkl").is_file()
    assert (out / "metadata.json").is_file()


def test_output_scores_metadata_aggregate():
    """Aggregate fold-mean == mean of the per-tag fold-means (reference
    test_output_scores_metadata), through a TransformedTargetRegressor."""
    model = {
        "gordo.machine.model.anomaly.diff.DiffBasedAnomalyDetector": {
            "require_thresholds": False,
            "scaler": "sklearn.preprocessing.MinMaxScaler",
            "base_estimator": {
                "sklearn.compose.TransformedTargetRegressor": {
                    "transformer": "sklearn.preprocessing.MinMaxScaler",
                    "regressor": {
                        "sklearn.pipeline.Pipeline": {
                            "steps": [
                                "sklearn.preprocessing.MinMaxScaler",
                                {
                                    "gordo.machine.model.models.KerasAutoEncoder": {
                                        "kind": "feedforward_hourglass",
                                        "compression_factor": 0.5,
                                        "encoding_layers": 1,
                                        "func": "tanh",
                                        "out_func": "linear",
                                        "epochs": 1,
                                    }
                                },
                            ]
                        }
                    },
                }
            },
        }
    }
    _, machine_out = ModelBuilder(make_machine(model=model)).build()
    scores = machine_out.metadata.build_metadata.model.cross_validation.scores
    for s in (
        "explained-variance-score",
        "r2-score",
        "mean-squared-error",
        "mean-absolute-error",
    ):
        per_tag = (scores[f"{s}-Tag-1"]["fold-mean"]
                   + scores[f"{s}-Tag-2"]["fold-mean"]) / 2
        assert per_tag == pytest.approx(scores[s]["fold-mean"])


@pytest.mark.parametrize(
    "metrics_",
    (
        ["sklearn.metrics.r2_score"],
        ["r2_score"],  # bare names resolve in sklearn.metrics
        None,
        ["sklearn.metrics.r2_score",
         "sklearn.metrics.explained_variance_score"],
    ),
)
def test_model_builder_metrics_list(metrics_):
    evaluation = {"cv_mode": "full_build"}
    if metrics_:
        evaluation["metrics"] = metrics_
    machine = make_machine(
        model={
            "sklearn.multioutput.MultiOutputRegressor": {
                "estimator": "sklearn.linear_model.LinearRegression"
            }
        },
        evaluation=evaluation,
    )
    _, machine_out = ModelBuilder(machine).build()
    expected = metrics_ or [
        "sklearn.metrics.explained_variance_score",
        "sklearn.metrics.r2_score",
        "sklearn.metrics.mean_squared_error",
        "sklearn.metrics.mean_absolute_error",
    ]
    scores = machine_out.metadata.build_metadata.model.cross_validation.scores
    assert all(
        m.split(".")[-1].replace("_", "-") in scores for m in expected
    )


@pytest.mark.parametrize(
    "cv",
    (
        {"sklearn.model_selection.TimeSeriesSplit": {
            "n_splits": 5, "max_train_size": 10}},
        {"sklearn.model_selection.ShuffleSplit": {
            "n_splits": 5, "random_state": 0}},
        None,
    ),
)
def test_n_splits_from_config(cv):
    """Arbitrary CV splitters configured via evaluation.cv are honored
    (reference test_n_splits_from_config) — fold count in metadata."""
    evaluation = {"cv_mode": "cross_val_only"}
    if cv:
        evaluation["cv"] = cv
    machine = make_machine(
        model={
            "sklearn.multioutput.MultiOutputRegressor": {
                "estimator": "sklearn.linear_model.LinearRegression"
            }
        },
        evaluation=evaluation,
    )
    _, machine_out = ModelBuilder(machine).build()
    scores = machine_out.metadata.build_metadata.model.cross_validation.scores
    n_expected = 5 if cv else 3
    folds = [k for k in scores["r2-score"] if k.startswith("fold-")]
    # fold-mean/std/max/min + fold-1..n
    assert len(folds) == n_expected + 4


def test_cv_mode_build_only(tmp_path):
    """build_only: model saved, no CV run, scores empty (reference
    tests/gordo/cli/test_cli.py::test_build_cv_mode_build_only)."""
    import json

    machine = make_machine(
        model=SKLEARN_MODEL, evaluation={"cv_mode": "build_only"}
    )
    ModelBuilder(machine).build(output_dir=str(tmp_path))
    md = json.load(open(tmp_path / "metadata.json"))
    cv = md["metadata"]["build_metadata"]["model"]["cross_validation"]
    assert cv["cv_duration_sec"] is None
    assert cv["scores"] == {}
    assert sorted(json.load(open(tmp_path / "info.json"))) == ["checksum"]
    assert (tmp_path / "model.pkl").is_file()
