import os
import sys
import tempfile

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

SENSORS = [f"tag-{i}" for i in range(4)]
GORDO_NAME = "machine-1"
SECOND_GORDO_NAME = "machine-2"
PROJECT = "gordo-test"
REVISION = "1577836800000"


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests requiring an AMD GPU (MI355X); skipped on CPU"
    )
    config.addinivalue_line("markers", "dockertest: tests requiring docker")


def pytest_collection_modifyitems(config, items):
    import torch

    if not torch.cuda.is_available():
        skip_gpu = pytest.mark.skip(reason="no GPU available")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip_gpu)


@pytest.fixture(scope="session")
def sensors():
    return SENSORS


@pytest.fixture(scope="session")
def gordo_name():
    return GORDO_NAME


@pytest.fixture(scope="session")
def second_gordo_name():
    return SECOND_GORDO_NAME


@pytest.fixture(scope="session")
def gordo_project():
    return PROJECT


@pytest.fixture(scope="session")
def gordo_revision():
    return REVISION


@pytest.fixture(scope="session")
def config_str(sensors):
    """Default 2-machine config used by the server integration tests
    (mirrors the reference's tests/conftest.py fixture shape)."""
    tag_block = "\n".join(f"        - {s}" for s in sensors)
    return f"""
machines:
  - dataset: |
      tags:
{tag_block}
      target_tag_list:
{tag_block}
      train_start_date: '2019-01-01T00:00:00+00:00'
      train_end_date: '2019-01-04T00:00:00+00:00'
      type: RandomDataset
    metadata: |
      information: Some sweet information about the model
    model: |
      gordo.machine.model.anomaly.diff.DiffBasedAnomalyDetector:
        require_thresholds: false
        base_estimator:
          sklearn.pipeline.Pipeline:
            steps:
            - sklearn.preprocessing.MinMaxScaler
            - gordo.machine.model.models.KerasAutoEncoder:
                kind: feedforward_hourglass
                epochs: 1
    name: {GORDO_NAME}
  - dataset: |
      tags:
{tag_block}
      target_tag_list:
{tag_block}
      train_start_date: '2019-01-01T00:00:00+00:00'
      train_end_date: '2019-01-04T00:00:00+00:00'
      type: RandomDataset
    metadata: |
      information: Some sweet information about the model
    model: |
      gordo.machine.model.anomaly.diff.DiffBasedAnomalyDetector:
        window: 10
        require_thresholds: false
        base_estimator:
          sklearn.pipeline.Pipeline:
            steps:
            - sklearn.preprocessing.MinMaxScaler
            - gordo.machine.model.models.KerasAutoEncoder:
                kind: feedforward_hourglass
                epochs: 1
    name: {SECOND_GORDO_NAME}
"""


@pytest.fixture(scope="session")
def model_collection_directory(gordo_revision):
    with tempfile.TemporaryDirectory() as tmp_dir:
        collection_dir = os.path.join(tmp_dir, gordo_revision)
        os.makedirs(collection_dir, exist_ok=True)
        yield collection_dir


@pytest.fixture(scope="session")
def trained_model_directories(model_collection_directory, config_str):
    """Train the default config through local_build and dump into a
    model-collection dir (the server-test backbone)."""
    from gordo_amd import serializer
    from gordo_amd.builder import local_build

    model_directories = {}
    for model, machine in local_build(config_str=config_str):
        model_dir = os.path.join(model_collection_directory, machine.name)
        os.makedirs(model_dir, exist_ok=True)
        serializer.dump(
            model, model_dir,
            metadata=__import__("json").loads(machine.to_json()),
            info={},
        )
        model_directories[machine.name] = model_dir
    yield model_directories


@pytest.fixture(scope="session")
def trained_model_directory(trained_model_directories, gordo_name):
    return trained_model_directories[gordo_name]


@pytest.fixture
def flask_app(model_collection_directory, trained_model_directories, gordo_project):
    os.environ["MODEL_COLLECTION_DIR"] = model_collection_directory
    from gordo_amd.server.server import build_app

    app = build_app()
    app.testing = True
    yield app


@pytest.fixture
def api_client(flask_app):
    return flask_app.test_client()


@pytest.fixture
def base_route(gordo_project, gordo_name):
    return f"/gordo/v0/{gordo_project}/{gordo_name}"
