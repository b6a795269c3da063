import pytest

from gordo_amd.core import RandomDataset, SensorTag, normalize_sensor_tag
from gordo_amd.machine import Machine, load_machine_config, load_model_config
from gordo_amd.machine.loader import MachineConfigException
from gordo_amd.machine.validators import ValidUrlString, fix_resource_limits


MODEL_DEF = {"sklearn.decomposition.PCA": {"n_components": 2}}
DATASET_DEF = {
    "type": "RandomDataset",
    "train_start_date": "2017-12-25 06:00:00Z",
    "train_end_date": "2017-12-30 06:00:00Z",
    "tag_list": ["Tag 1", "Tag 2"],
}


def _machine(**overrides):
    cfg = dict(name="test-machine", model=MODEL_DEF, dataset=dict(DATASET_DEF))
    cfg.update(overrides)
    return Machine.from_config(cfg, project_name="test-proj")


def test_from_config_roundtrip():
    m = _machine()
    assert m.name == "test-machine"
    assert m.project_name == "test-proj"
    assert m.evaluation["cv_mode"] == "full_build"
    d = m.to_dict()
    m2 = Machine.from_dict(d)
    assert m2 == m
    # yaml/json roundtrips parse
    assert "test-machine" in m.to_yaml()
    assert "test-machine" in m.to_json()


def test_globals_merge_precedence():
    config_globals = {
        "model": {"sklearn.decomposition.PCA": {"n_components": 5}},
        "runtime": {"server": {"resources": {"requests": {"memory": 1}}}},
        "evaluation": {"cv_mode": "cross_val_only", "seed": 7},
        "dataset": {"resolution": "1h"},
    }
    m = Machine.from_config(
        {
            "name": "m",
            "dataset": dict(DATASET_DEF),
            "model": MODEL_DEF,
            "evaluation": {"cv_mode": "full_build"},
        },
        project_name="p",
        config_globals=config_globals,
    )
    # machine model wins over globals model
    assert m.model == MODEL_DEF
    # machine evaluation overrides globals, but globals fill gaps
    assert m.evaluation["cv_mode"] == "full_build"
    assert m.evaluation["seed"] == 7
    # reference quirk: globals dataset overlays machine's
    assert m.dataset.resolution == "1h"


def test_invalid_name_rejected():
    with pytest.raises(ValueError):
        _machine(name="Bad_Name")
    with pytest.raises(ValueError):
        _machine(name="x" * 64)


def test_invalid_model_rejected():
    with pytest.raises(ValueError):
        _machine(model={"not.a.real.Class": {}})


def test_nested_yaml_fields():
    cfg = load_machine_config(
        {
            "name": "m",
            "model": "sklearn.decomposition.PCA:\n  n_components: 2\n",
            "dataset": "type: RandomDataset\ntag_list: [a]\n"
                       "train_start_date: 2019-01-01T00:00:00Z\n"
                       "train_end_date: 2019-01-02T00:00:00Z\n",
        }
    )
    assert cfg["model"] == {"sklearn.decomposition.PCA": {"n_components": 2}}
    assert cfg["dataset"]["type"] == "RandomDataset"


def test_load_model_config_requires_model():
    with pytest.raises(MachineConfigException):
        load_model_config({"name": "m"})
    with pytest.raises(MachineConfigException):
        load_machine_config({"model": MODEL_DEF})


def test_valid_url_string():
    assert ValidUrlString.valid_url_string("good-name-01")
    assert not ValidUrlString.valid_url_string("Bad Name")
    assert not ValidUrlString.valid_url_string("-leading")


def test_fix_resource_limits():
    res = fix_resource_limits(
        {"requests": {"memory": 10, "cpu": 5}, "limits": {"memory": 4, "cpu": 9}}
    )
    assert res["limits"]["memory"] == 10
    assert res["limits"]["cpu"] == 9


def test_sensor_tag_normalization():
    assert normalize_sensor_tag("a").name == "a"
    assert normalize_sensor_tag({"name": "a", "asset": "x"}).asset == "x"
    assert normalize_sensor_tag(["a", "x"]).asset == "x"
    assert normalize_sensor_tag(SensorTag("a", "x")) == SensorTag("a", "x")


def test_dataset_get_data_shapes():
    ds = RandomDataset(
        train_start_date="2019-01-01T00:00:00Z",
        train_end_date="2019-01-02T00:00:00Z",
        tag_list=["a", "b", "c"],
        target_tag_list=["a", "b"],
    )
    X, y = ds.get_data()
    assert list(X.columns) == ["a", "b", "c"]
    assert list(y.columns) == ["a", "b"]
    assert len(X) == len(y) > 0
    meta = ds.get_metadata()
    assert meta["resolution"] == "10min"


def test_dataset_requires_timezone():
    from gordo_amd.core.exceptions import ConfigException

    with pytest.raises(ConfigException):
        RandomDataset(
            train_start_date="2019-01-01T00:00:00",
            train_end_date="2019-01-02T00:00:00",
            tag_list=["a"],
        )


def test_load_globals_config_nested_yaml():
    """Every machine-level field given as a YAML string parses (reference
    tests/gordo/machine/test_loader.py::test_load_globals_config)."""
    import yaml as _yaml

    from gordo_amd.machine.loader import (
        load_globals_config,
        load_machine_config,
    )

    config = _yaml.safe_load(
        """
dataset: |
  tags:
  - tag1
  - tag2
  train_end_date: '2022-01-15T00:00:00+00:00'
  train_start_date: '2021-12-25T00:00:00+00:00'
model: |
  gordo.machine.model.anomaly.diff.DiffBasedAnomalyDetector:
    base_estimator:
      sklearn.pipeline.Pipeline:
        steps:
          - sklearn.preprocessing.MinMaxScaler
          - gordo.machine.model.models.KerasAutoEncoder:
              kind: feedforward_hourglass
runtime: |
  builder:
    resources:
      requests:
        memory: 1000
metadata: |
  key1: value1
evaluation: |
  cv_mode: no_cv
"""
    )
    out = load_globals_config(dict(config))
    assert out["dataset"]["tags"] == ["tag1", "tag2"]
    assert out["runtime"]["builder"]["resources"]["requests"]["memory"] == 1000
    assert out["metadata"] == {"key1": "value1"}
    assert out["evaluation"] == {"cv_mode": "no_cv"}
    model_key = next(iter(out["model"]))
    assert model_key.endswith("DiffBasedAnomalyDetector")

    # machine config: same parsing + name survives
    mc = load_machine_config({"name": "m1", **config})
    assert mc["name"] == "m1"
    assert mc["evaluation"] == {"cv_mode": "no_cv"}


def test_load_globals_config_bad_yaml_raises():
    from gordo_amd.machine.loader import load_globals_config

    with pytest.raises(Exception) as exc:
        load_globals_config({"model": ": not :\n - valid yaml ["})
    assert "model" in str(exc.value)


def test_machine_report_runs_reporters(tmp_path):
    """machine.report() instantiates & runs every runtime reporter
    (reference tests/gordo/machine/test_machine.py::test_builder_with_reporter,
    without the live postgres)."""
    from gordo_amd.machine import Machine
    from gordo_amd.reporters.base import BaseReporter

    reported = []

    class _CollectingReporter(BaseReporter):
        def report(self, machine):
            reported.append(machine.name)

    m = Machine.from_config(
        {
            "name": "report-m",
            "model": {"sklearn.decomposition.PCA": {"n_components": 2}},
            "dataset": {
                "type": "RandomDataset",
                "tag_list": ["a", "b", "c"],
                "train_start_date": "2019-01-01T00:00:00Z",
                "train_end_date": "2019-01-02T00:00:00Z",
            },
            "runtime": {"reporters": []},
        },
        project_name="p",
    )
    m.runtime["reporters"] = [_CollectingReporter()]
    m.report()
    assert reported == ["report-m"]


def test_sinewave_load_frame_bit_identical():
    """The provider's whole-frame fast path produces bit-identical data
    to the per-series join path."""
    import gordo_amd.core.data_providers as dpm
    from gordo_amd.core import SineWaveDataset

    ds = SineWaveDataset(
        tag_list=[f"s-{j}" for j in range(5)],
        train_start_date="2019-01-01T00:00:00Z",
        train_end_date="2019-01-02T00:00:00Z",
    )
    X1, y1 = ds.get_data()
    orig = dpm.SineWaveDataProvider.load_frame
    dpm.SineWaveDataProvider.load_frame = dpm.DataProvider.load_frame
    try:
        X2, y2 = ds.get_data()
    finally:
        dpm.SineWaveDataProvider.load_frame = orig
    assert (X1.values == X2.values).all()
    assert list(X1.columns) == list(X2.columns)
    assert (X1.index == X2.index).all()
    assert (y1.values == y2.values).all()


def test_machine_equality_and_repr():
    from gordo_amd.machine import Machine

    cfg = {
        "name": "eq-m",
        "model": {"sklearn.decomposition.PCA": {"n_components": 2}},
        "dataset": {
            "type": "RandomDataset",
            "tag_list": ["a", "b", "c"],
            "train_start_date": "2019-01-01T00:00:00Z",
            "train_end_date": "2019-01-02T00:00:00Z",
        },
    }
    m1 = Machine.from_config(cfg, project_name="p")
    m2 = Machine.from_config(cfg, project_name="p")
    assert m1 == m2
    m3 = Machine.from_config({**cfg, "name": "eq-n"}, project_name="p")
    assert m1 != m3
    assert "eq-m" in repr(m1) or "eq-m" in str(m1.to_dict())
