import numpy as np
import pytest
from sklearn.decomposition import PCA
from sklearn.pipeline import Pipeline, FeatureUnion
from sklearn.preprocessing import MinMaxScaler

from gordo_amd import serializer
from gordo_amd.machine.model.models import KerasAutoEncoder


def test_from_definition_basic_pipeline():
    definition = {
        "sklearn.pipeline.Pipeline": {
            "steps": [
                "sklearn.preprocessing.MinMaxScaler",
                {"sklearn.decomposition.PCA": {"n_components": 3}},
            ]
        }
    }
    pipe = serializer.from_definition(definition)
    assert isinstance(pipe, Pipeline)
    assert isinstance(pipe.steps[0][1], MinMaxScaler)
    assert isinstance(pipe.steps[1][1], PCA)
    assert pipe.steps[1][1].n_components == 3


def test_from_definition_gordo_alias():
    """Reference configs use gordo.* import paths; they must resolve."""
    definition = {
        "gordo.machine.model.models.KerasAutoEncoder": {
            "kind": "feedforward_hourglass"
        }
    }
    model = serializer.from_definition(definition)
    assert isinstance(model, KerasAutoEncoder)
    assert model.kind == "feedforward_hourglass"


def test_from_definition_feature_union():
    definition = {
        "sklearn.pipeline.FeatureUnion": {
            "transformer_list": [
                {"sklearn.decomposition.PCA": {"n_components": 2}},
                "sklearn.preprocessing.MinMaxScaler",
            ]
        }
    }
    union = serializer.from_definition(definition)
    assert isinstance(union, FeatureUnion)
    assert len(union.transformer_list) == 2


def test_round_trip_into_from():
    pipe = Pipeline(
        [("mms", MinMaxScaler()), ("pca", PCA(n_components=2))]
    )
    definition = serializer.into_definition(pipe)
    rebuilt = serializer.from_definition(definition)
    assert isinstance(rebuilt, Pipeline)
    assert rebuilt.steps[1][1].n_components == 2
    # double round trip is stable
    assert serializer.into_definition(rebuilt) == definition


def test_round_trip_nested_detector():
    from gordo_amd.machine.model.anomaly.diff import DiffBasedAnomalyDetector

    definition = {
        "gordo_amd.machine.model.anomaly.diff.DiffBasedAnomalyDetector": {
            "require_thresholds": False,
            "base_estimator": {
                "sklearn.pipeline.Pipeline": {
                    "steps": [
                        "sklearn.preprocessing.MinMaxScaler",
                        {
                            "gordo_amd.machine.model.models.KerasAutoEncoder": {
                                "kind": "feedforward_hourglass"
                            }
                        },
                    ]
                }
            },
        }
    }
    model = serializer.from_definition(definition)
    assert isinstance(model, DiffBasedAnomalyDetector)
    back = serializer.into_definition(model)
    rebuilt = serializer.from_definition(back)
    assert isinstance(rebuilt, DiffBasedAnomalyDetector)
    assert isinstance(rebuilt.base_estimator, Pipeline)


def test_dump_load_layout(tmp_path):
    model = Pipeline([("mms", MinMaxScaler())])
    model.fit(np.random.rand(10, 2))
    serializer.dump(
        model, str(tmp_path), metadata={"name": "m"}, info={}
    )
    assert (tmp_path / "model.pkl").is_file()
    assert (tmp_path / "metadata.json").is_file()
    assert (tmp_path / "info.json").is_file()
    loaded = serializer.load(str(tmp_path))
    assert isinstance(loaded, Pipeline)
    meta = serializer.load_metadata(str(tmp_path))
    assert meta["name"] == "m"
    info = serializer.load_info(str(tmp_path))
    assert "checksum" in info


def test_load_metadata_searches_parent(tmp_path):
    sub = tmp_path / "sub"
    sub.mkdir()
    (tmp_path / "metadata.json").write_text('{"x": 1}')
    assert serializer.load_metadata(str(sub)) == {"x": 1}


def test_dumps_loads_bytes():
    model = MinMaxScaler()
    blob = serializer.dumps(model)
    assert isinstance(blob, bytes)
    assert isinstance(serializer.loads(blob), MinMaxScaler)


def test_from_definition_rejects_bad_path():
    with pytest.raises((ImportError, ValueError)):
        serializer.from_definition({"no.such.module.Klass": {}})


@pytest.mark.parametrize(
    "definition",
    [
        """
sklearn.multioutput.MultiOutputRegressor:
  estimator: sklearn.ensemble.RandomForestRegressor
""",
        """
sklearn.multioutput.MultiOutputRegressor:
  estimator:
    sklearn.ensemble.RandomForestRegressor:
      n_estimators: 20
""",
        """
sklearn.multioutput.MultiOutputRegressor:
  estimator:
    sklearn.pipeline.Pipeline:
      steps:
        - sklearn.ensemble.RandomForestRegressor:
            n_estimators: 20
""",
        """
sklearn.multioutput.MultiOutputRegressor:
  estimator:
    sklearn.pipeline.Pipeline:
      steps:
        - sklearn.cluster.FeatureAgglomeration:
            n_clusters: 2
            pooling_func: numpy.mean
        - sklearn.linear_model.LinearRegression
""",
    ],
)
def test_models_as_parameters(definition):
    """Estimator-valued params in every spelling: bare class string,
    class-with-kwargs dict, nested Pipeline, callable param
    (numpy.mean) — reference test_serializer_from_definition.py:27-70."""
    import numpy as np
    from sklearn.multioutput import MultiOutputRegressor

    import yaml

    model = serializer.from_definition(yaml.safe_load(definition))
    assert isinstance(model, MultiOutputRegressor)
    X, y = np.random.random((10, 10)), np.random.random((10, 2))
    model.fit(X, y)
    model.predict(X)


def test_into_definition_captures_kwargs():
    """into_definition records non-default init kwargs so the definition
    round-trips (reference test_captures_kwarg_to_init)."""
    from sklearn.decomposition import PCA
    from sklearn.pipeline import Pipeline

    pipe = Pipeline([("pca", PCA(n_components=3, whiten=True))])
    definition = serializer.into_definition(pipe)
    step = definition["sklearn.pipeline.Pipeline"]["steps"][0]
    key = next(k for k in step if k.endswith(".PCA"))
    params = step[key]
    assert params["n_components"] == 3
    assert params["whiten"] is True


def test_info_json_checksum_content(tmp_path):
    """info.json carries a checksum over the pickled model (the
    reference's cache-integrity breadcrumb)."""
    from sklearn.preprocessing import MinMaxScaler

    serializer.dump(
        MinMaxScaler(), str(tmp_path),
        info={"checksum": "abc123"},
    )
    info = serializer.load_info(str(tmp_path))
    assert info == {"checksum": "abc123"}
