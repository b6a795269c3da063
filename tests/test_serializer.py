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
