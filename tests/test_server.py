import json
import os
import sys

import numpy as np
import pandas as pd
import pytest

from gordo_amd.server import utils as server_utils


@pytest.fixture
def X(sensors):
    rng = np.random.default_rng(1)
    index = pd.date_range("2019-01-01", periods=50, freq="10min", tz="UTC")
    return pd.DataFrame(
        rng.random((50, len(sensors))), columns=sensors, index=index
    )


def _post_json(client, url, X, y=None, **params):
    payload = {"X": server_utils.dataframe_to_dict(X)}
    if y is not None:
        payload["y"] = server_utils.dataframe_to_dict(y)
    return client.post(url, json=payload, query_string=params)


def test_healthcheck(api_client):
    resp = api_client.get("/healthcheck")
    assert resp.status_code == 200


def test_server_version(api_client):
    resp = api_client.get("/server-version")
    assert resp.status_code == 200
    assert "version" in resp.json


def test_metadata_endpoint(api_client, base_route, gordo_name):
    resp = api_client.get(f"{base_route}/metadata")
    assert resp.status_code == 200
    body = resp.json
    assert body["metadata"]["name"] == gordo_name
    assert "gordo-server-version" in body
    assert "revision" in body
    # /healthcheck alias
    resp2 = api_client.get(f"{base_route}/healthcheck")
    assert resp2.status_code == 200


def test_models_list(api_client, gordo_project, gordo_name, second_gordo_name):
    resp = api_client.get(f"/gordo/v0/{gordo_project}/models")
    assert resp.status_code == 200
    assert set(resp.json["models"]) >= {gordo_name, second_gordo_name}


def test_revisions(api_client, gordo_project, gordo_revision):
    resp = api_client.get(f"/gordo/v0/{gordo_project}/revisions")
    assert resp.status_code == 200
    assert resp.json["latest"] == gordo_revision
    assert gordo_revision in resp.json["available-revisions"]


def test_expected_models(api_client, gordo_project):
    resp = api_client.get(f"/gordo/v0/{gordo_project}/expected-models")
    assert resp.status_code == 200
    assert "expected-models" in resp.json


def test_prediction_json(api_client, base_route, X):
    resp = _post_json(api_client, f"{base_route}/prediction", X)
    assert resp.status_code == 200
    data = resp.json["data"]
    assert "model-input" in data and "model-output" in data
    assert len(data["model-output"]) == X.shape[1]
    assert resp.headers.get("revision")


def test_prediction_parquet(api_client, base_route, X):
    import io

    blob = server_utils.dataframe_into_parquet_bytes(X)
    resp = api_client.post(
        f"{base_route}/prediction?format=parquet",
        data={"X": (io.BytesIO(blob), "X")},
        content_type="multipart/form-data",
    )
    assert resp.status_code == 200
    frame = server_utils.dataframe_from_parquet_bytes(resp.data)
    assert "model-output" in {c[0] for c in frame.columns}
    assert len(frame) == len(X)


def test_prediction_without_x(api_client, base_route):
    resp = api_client.post(f"{base_route}/prediction", json={})
    assert resp.status_code == 400


def test_prediction_wrong_columns(api_client, base_route, X):
    bad = X.iloc[:, :2]
    resp = _post_json(api_client, f"{base_route}/prediction", bad)
    assert resp.status_code == 400


def test_prediction_unknown_model(api_client, gordo_project, X):
    resp = _post_json(
        api_client, f"/gordo/v0/{gordo_project}/no-such-model/prediction", X
    )
    assert resp.status_code == 404


def test_anomaly_prediction(api_client, base_route, X):
    resp = _post_json(api_client, f"{base_route}/anomaly/prediction", X, y=X)
    assert resp.status_code == 200
    data = resp.json["data"]
    for key in (
        "model-input", "model-output",
        "tag-anomaly-scaled", "total-anomaly-scaled",
        "tag-anomaly-unscaled", "total-anomaly-unscaled",
    ):
        assert key in data, key
    # smoothed columns dropped by default
    assert not any(k.startswith("smooth-") for k in data)
    assert "time-seconds" in resp.json


def test_anomaly_prediction_all_columns(
    api_client, gordo_project, second_gordo_name, X
):
    url = f"/gordo/v0/{gordo_project}/{second_gordo_name}/anomaly/prediction"
    resp = _post_json(api_client, url, X, y=X, all_columns="true")
    assert resp.status_code == 200
    assert any(k.startswith("smooth-") for k in resp.json["data"])


def test_anomaly_requires_y(api_client, base_route, X):
    resp = _post_json(api_client, f"{base_route}/anomaly/prediction", X)
    assert resp.status_code == 400


def test_download_model(api_client, base_route):
    resp = api_client.get(f"{base_route}/download-model")
    assert resp.status_code == 200
    from gordo_amd import serializer

    model = serializer.loads(resp.data)
    assert hasattr(model, "anomaly")


def test_revision_query(api_client, base_route, gordo_revision, X):
    resp = _post_json(
        api_client, f"{base_route}/prediction", X, revision=gordo_revision
    )
    assert resp.status_code == 200
    assert resp.headers["revision"] == gordo_revision


def test_unknown_revision_410(api_client, base_route, X):
    resp = _post_json(
        api_client, f"{base_route}/prediction", X, revision="123"
    )
    assert resp.status_code == 410
    resp = _post_json(
        api_client, f"{base_route}/prediction", X, revision="notanumber"
    )
    assert resp.status_code == 410


def test_dataframe_dict_roundtrip(X):
    d = server_utils.dataframe_to_dict(X)
    df = server_utils.dataframe_from_dict(d)
    np.testing.assert_allclose(df.values, X.values, rtol=1e-9)


def test_parquet_roundtrip(X):
    blob = server_utils.dataframe_into_parquet_bytes(X)
    df = server_utils.dataframe_from_parquet_bytes(blob)
    np.testing.assert_allclose(df.values, X.values)


def test_proxy_path_rewrite(flask_app):
    # Envoy-prefixed request should reach the route
    resp = flask_app.test_client().get(
        "/healthcheck",
        environ_overrides={
            "HTTP_X_ENVOY_ORIGINAL_PATH": "/gordo/v0/proj/healthcheck"
        },
    )
    assert resp.status_code == 200


def test_server_timing_header(api_client):
    resp = api_client.get("/healthcheck")
    assert "Server-Timing" in resp.headers
    assert "request_walltime_s" in resp.headers["Server-Timing"]


def test_invalid_gordo_name_422(api_client, gordo_project, X):
    resp = _post_json(
        api_client, f"/gordo/v0/{gordo_project}/bad_name!/prediction", X
    )
    assert resp.status_code == 422


def test_metadata_404_unknown_model(api_client, gordo_project):
    resp = api_client.get(f"/gordo/v0/{gordo_project}/no-such/metadata")
    assert resp.status_code == 404


def test_delete_revision_endpoint(
    flask_app, gordo_project, gordo_name, model_collection_directory
):
    import shutil

    # create an OLD revision alongside the current one
    parent = os.path.dirname(model_collection_directory)
    old_rev = os.path.join(parent, "1000000000000")
    src = os.path.join(model_collection_directory, gordo_name)
    shutil.copytree(src, os.path.join(old_rev, gordo_name),
                    dirs_exist_ok=True)
    client = flask_app.test_client()

    # the old revision serves
    resp = client.get(
        f"/gordo/v0/{gordo_project}/{gordo_name}/metadata",
        query_string={"revision": "1000000000000"},
    )
    assert resp.status_code == 200

    # deleting the CURRENT revision is refused
    cur = os.path.basename(model_collection_directory)
    resp = client.delete(
        f"/gordo/v0/{gordo_project}/{gordo_name}/revision/{cur}"
    )
    assert resp.status_code == 409

    # non-numeric revision refused
    resp = client.delete(
        f"/gordo/v0/{gordo_project}/{gordo_name}/revision/abc"
    )
    assert resp.status_code == 422

    # deleting the old revision works and removes it from disk
    resp = client.delete(
        f"/gordo/v0/{gordo_project}/{gordo_name}/revision/1000000000000"
    )
    assert resp.status_code == 200
    assert not os.path.exists(old_rev)


def test_with_prometheus_enabled(model_collection_directory,
                                 trained_model_directories):
    from prometheus_client.registry import CollectorRegistry

    from gordo_amd.server.server import build_app

    os.environ["MODEL_COLLECTION_DIR"] = model_collection_directory
    app = build_app(
        config={"ENABLE_PROMETHEUS": True, "PROJECT": "prom-proj"},
        prometheus_registry=CollectorRegistry(),
    )
    app.testing = True
    resp = app.test_client().get("/healthcheck")
    assert resp.status_code == 200


def test_run_cmd():
    import subprocess

    from gordo_amd.server.server import run_cmd

    run_cmd([sys.executable, "-c", "print('ok')"])
    with pytest.raises(subprocess.CalledProcessError):
        run_cmd([sys.executable, "-c", "raise SystemExit(3)"])


@pytest.mark.parametrize(
    "expect_multi_lvl, data",
    [
        (False, {"col1": [0, 1, 2, 3], "col2": [0, 1, 2, 3]}),
        (True, {("ft1", "col1"): [0, 1], ("ft1", "col2"): [0, 1]}),
        (True, {"ft1": {"col1": [0, 1, 2]}, "ft2": {"col1": [0, 1, 2]}}),
        (False, [[0, 1, 2], [0, 1, 2]]),
    ],
)
def test_dataframe_from_dict_shapes(expect_multi_lvl, data):
    """Raw payload shapes map deterministically to flat vs MultiIndex
    columns (reference tests/gordo/server/test_utils.py:78-88)."""
    df = server_utils.dataframe_from_dict(data)
    assert isinstance(df.columns, pd.MultiIndex) == expect_multi_lvl


def test_multilevel_roundtrip():
    df = pd.DataFrame(
        np.random.default_rng(0).random((10, 4)),
        columns=pd.MultiIndex.from_product((("f1", "f2"), ("c1", "c2"))),
        index=pd.date_range("2016-01-01", "2016-02-01", periods=10),
    )
    c = server_utils.dataframe_from_dict(server_utils.dataframe_to_dict(df))
    assert isinstance(c.columns, pd.MultiIndex)
    assert df.columns.tolist() == c.columns.tolist()
    assert df.index.tolist() == c.index.tolist()
    np.testing.assert_allclose(df.values, c.values)
    # the input frame's own DatetimeIndex is not mutated
    assert isinstance(df.index, pd.DatetimeIndex)


@pytest.mark.parametrize(
    "index",
    (
        list(range(10)),
        [str(i) for i in (3, 1, 4, 0, 9, 2, 8, 5, 7, 6)],
        pd.date_range("2020-01-01", "2020-01-02", periods=10),
        pd.date_range("2020-01-01", "2020-01-02", periods=10).astype(
            str
        ).tolist()[::-1],
    ),
)
def test_dataframe_from_dict_ordering(index):
    """from_dict parses string indexes as datetimes or ints and sorts
    ascending (reference test_utils.py:109-131)."""
    import dateutil.parser

    df = pd.DataFrame(np.random.default_rng(1).random((10, 5)))
    df.index = index
    original = df.copy()
    if isinstance(original.index[0], str):
        try:
            original.index = original.index.map(dateutil.parser.isoparse)
        except ValueError:
            original.index = original.index.map(int)
    original = original.sort_index()
    out = server_utils.dataframe_from_dict(server_utils.dataframe_to_dict(df))
    assert list(out.index) == list(original.index)
    np.testing.assert_allclose(out.values, original.values)


def test_fastjson_frame_encoder_equivalence():
    """The C++ response encoder is byte-identical to
    json.dumps(dataframe_to_dict(df)) on response-shaped frames,
    including NaN/inf, integral floats and escaped keys."""
    fj = pytest.importorskip("gordo_amd.server._gordo_fastjson")
    import json

    rng = np.random.default_rng(3)
    df = pd.DataFrame(
        rng.random((100, 100)),
        columns=pd.MultiIndex.from_product(
            (("model-input", "model-output"), [f"t{i}" for i in range(50)])
        ),
        index=pd.date_range("2020-01-01", periods=100, freq="10min"),
    )
    df.iloc[0, 0] = float("nan")
    df.iloc[1, 1] = float("inf")
    df.iloc[2, 2] = -float("inf")
    df.iloc[3, 3] = 1.0
    fast = fj.encode_frame(
        df.index.astype(str).tolist(),
        [c[0] for c in df.columns],
        [c[1] for c in df.columns],
        df.values,
    )
    ref = json.dumps(server_utils.dataframe_to_dict(df)).encode()
    assert fast == ref

    df2 = pd.DataFrame(
        [[1.5]],
        columns=pd.MultiIndex.from_tuples([('a"b', "c\\d\te")]),
        index=["i\n1"],
    )
    fast2 = fj.encode_frame(
        ["i\n1"], ['a"b'], ["c\\d\te"], df2.values
    )
    assert fast2 == json.dumps(server_utils.dataframe_to_dict(df2)).encode()

    # non-ascii keys are refused (callers fall back to the python codec)
    with pytest.raises(ValueError):
        fj.encode_frame(["ø"], ["a"], ["b"], np.zeros((1, 1)))


def test_prediction_json_fast_path_matches(api_client, base_route, X):
    """The endpoint response through frame_json_response parses to the
    same structure as the python codec path."""
    resp = _post_json(api_client, f"{base_route}/prediction", X)
    assert resp.status_code == 200
    assert resp.mimetype == "application/json"
    data = resp.json["data"]
    frame = server_utils.dataframe_from_dict(data)
    assert len(frame) == len(X)


def test_revisions_echo_requested_revision(
    api_client, base_route, gordo_project, gordo_revision,
    model_collection_directory,
):
    """/revisions carries latest + the revision this request resolved
    (reference test_gordo_server.py::test_list_revisions keys)."""
    resp = api_client.get(f"/gordo/v0/{gordo_project}/revisions")
    assert set(resp.json.keys()) == {
        "latest", "available-revisions", "revision"
    }
    assert resp.json["revision"] == gordo_revision

    # a second on-disk revision can be requested explicitly
    import shutil

    parent = os.path.dirname(model_collection_directory)
    other = os.path.join(parent, "1000000000001")
    shutil.copytree(model_collection_directory, other, dirs_exist_ok=True)
    try:
        resp = api_client.get(
            f"/gordo/v0/{gordo_project}/revisions",
            query_string={"revision": "1000000000001"},
        )
        assert resp.json["revision"] == "1000000000001"
        assert resp.json["latest"] == gordo_revision
    finally:
        shutil.rmtree(other)


def test_models_list_missing_collection_dir():
    from gordo_amd.server.server import build_app

    os.environ["MODEL_COLLECTION_DIR"] = os.path.join(
        "does", "not", "exist", "1"
    )
    app = build_app(config={"ENABLE_PROMETHEUS": False})
    app.testing = True
    resp = app.test_client().get("/gordo/v0/any-proj/models")
    assert resp.status_code == 200
    assert resp.json["models"] == []


def test_prometheus_ignores_healthcheck(model_collection_directory,
                                        trained_model_directories):
    from prometheus_client.registry import CollectorRegistry

    from gordo_amd.server.server import build_app

    os.environ["MODEL_COLLECTION_DIR"] = model_collection_directory
    registry = CollectorRegistry()
    app = build_app(
        config={"ENABLE_PROMETHEUS": True, "PROJECT": "prom-proj2"},
        prometheus_registry=registry,
    )
    app.testing = True
    client = app.test_client()
    client.get("/healthcheck")
    client.get("/server-version")
    # /healthcheck not sampled; /server-version is
    names = {
        (s.labels.get("path"),)
        for metric in registry.collect()
        for s in metric.samples
        if metric.name == "gordo_server_requests"
    }
    assert ("/healthcheck",) not in names
    assert ("/server-version",) in names


def test_frame_json_response_nonadjacent_families():
    """Interleaved top-level column families fall back to the python
    codec rather than emitting duplicate JSON keys."""
    from gordo_amd.server.utils import frame_json_response
    from flask import Flask

    df = pd.DataFrame(
        [[1.0, 2.0, 3.0]],
        columns=pd.MultiIndex.from_tuples(
            [("a", "x"), ("b", "x"), ("a", "y")]
        ),
        index=["i"],
    )
    app = Flask("t")
    with app.test_request_context():
        resp = frame_json_response({}, df)
    data = json.loads(resp.get_data())["data"]
    assert data["a"] == {"x": {"i": 1.0}, "y": {"i": 3.0}}
    assert data["b"] == {"x": {"i": 2.0}}


def test_models_list_ignores_files(api_client, gordo_project,
                                   model_collection_directory):
    stray = os.path.join(model_collection_directory, ".fleet-status.json")
    with open(stray, "w") as f:
        f.write("{}")
    try:
        resp = api_client.get(f"/gordo/v0/{gordo_project}/models")
        assert ".fleet-status.json" not in resp.json["models"]
    finally:
        os.unlink(stray)


def test_run_server_gunicorn_argv(monkeypatch):
    """run_server builds the gunicorn command line of the reference
    (test_gordo_server.py::test_run_server_gthread/gevent shape) when
    gunicorn is importable; werkzeug fallback otherwise is exercised by
    the import guard."""
    import types

    from gordo_amd.server import server as srv

    calls = []
    monkeypatch.setattr(srv, "run_cmd", lambda cmd: calls.append(cmd))
    # fake out the gunicorn import so the argv path runs in this image
    monkeypatch.setitem(sys.modules, "gunicorn", types.ModuleType("gunicorn"))

    srv.run_server(
        "127.0.0.1", 9000, 2, "debug",
        worker_connections=50, threads=8, worker_class="gthread",
    )
    assert calls[-1] == [
        "gunicorn",
        "--bind", "127.0.0.1:9000",
        "--log-level", "debug",
        "--error-logfile", "-",
        "--access-logfile", "-",
        "--worker-tmp-dir", "/dev/shm",
        "--worker-class", "gthread",
        "--workers", "2",
        "--threads", "8",
        "gordo_amd.server.server:build_app()",
    ]

    srv.run_server(
        "127.0.0.1", 9000, 2, "debug",
        worker_connections=50, threads=8, worker_class="gevent",
    )
    assert "--worker-connections" in calls[-1]
    assert "--threads" not in calls[-1]


def test_machine_encoders():
    """Machine JSON encoder handles datetime + SensorTag; YAML dumper
    renders nested configs as | multiline blocks (reference
    machine/encoders.py)."""
    import datetime

    import yaml

    from gordo_amd.machine import Machine

    m = Machine.from_config(
        {
            "name": "enc-m",
            "model": {"sklearn.decomposition.PCA": {"n_components": 2}},
            "dataset": {
                "type": "RandomDataset",
                "tag_list": ["a", "b", "c"],
                "train_start_date": "2019-01-01T00:00:00Z",
                "train_end_date": "2019-01-02T00:00:00Z",
            },
        },
        project_name="p",
    )
    as_json = m.to_json()
    parsed = json.loads(as_json)
    # datetimes serialized as ISO strings; tags as JSON-able values
    start = parsed["dataset"]["train_start_date"]
    assert datetime.datetime.fromisoformat(start.replace("Z", "+00:00"))
    yaml_str = m.to_yaml()
    assert yaml.safe_load(yaml_str)["name"] == "enc-m"


def test_serving_device_hash(monkeypatch):
    """GORDO_SERVER_GPUS pins each model to a stable device index."""
    import types

    from gordo_amd.server import utils as su

    fake_torch = types.SimpleNamespace(
        cuda=types.SimpleNamespace(
            is_available=lambda: True, device_count=lambda: 8
        )
    )
    monkeypatch.setitem(sys.modules, "torch", fake_torch)
    monkeypatch.setenv("GORDO_SERVER_GPUS", "auto")
    d1 = su._serving_device_for("model-a")
    assert d1 == su._serving_device_for("model-a")  # stable
    assert d1.startswith("cuda:")
    devices = {su._serving_device_for(f"m-{i}") for i in range(64)}
    assert len(devices) > 1  # spreads across GPUs

    monkeypatch.setenv("GORDO_SERVER_GPUS", "1")
    assert su._serving_device_for("model-a") is None
    monkeypatch.delenv("GORDO_SERVER_GPUS")
    assert su._serving_device_for("model-a") is None


def test_set_serving_device_cpu_roundtrip():
    """set_serving_device('cpu') keeps inference working (the GPU path
    differs only in the device string)."""
    from gordo_amd.machine.model.models import KerasAutoEncoder

    X = np.random.RandomState(0).random((40, 6))
    model = KerasAutoEncoder(kind="feedforward_hourglass", epochs=1)
    model.fit(X)
    before = model.predict(X)
    model.set_serving_device("cpu")
    after = model.predict(X)
    np.testing.assert_allclose(before, after, rtol=1e-5, atol=1e-6)


def test_prediction_model_error_returns_400(
    flask_app, gordo_project, model_collection_directory, sensors
):
    """A model that raises during predict/transform maps to a 400 with
    an error body (reference base.py:30-113 error branches)."""
    import shutil

    from sklearn.preprocessing import FunctionTransformer

    from gordo_amd import serializer

    src = os.path.join(model_collection_directory, "machine-1")
    broken = os.path.join(model_collection_directory, "broken-model")
    shutil.copytree(src, broken)
    # transform = matrix inverse of a non-square frame -> LinAlgError
    # (a ValueError subclass), exercising the 400 branch
    serializer.dump(FunctionTransformer(np.linalg.inv), broken)

    client = flask_app.test_client()
    X = pd.DataFrame(
        np.random.default_rng(0).random((10, len(sensors))),
        columns=sensors,
        index=pd.date_range("2019-01-01", periods=10, freq="10min", tz="UTC"),
    )
    try:
        resp = client.post(
            f"/gordo/v0/{gordo_project}/broken-model/prediction",
            json={"X": server_utils.dataframe_to_dict(X)},
        )
        assert resp.status_code == 400
        assert "error" in resp.json
    finally:
        shutil.rmtree(broken)
        server_utils.load_model.cache_clear()


def test_anomaly_on_non_detector_422(
    flask_app, gordo_project, model_collection_directory, sensors
):
    """/anomaly/prediction on a model without .anomaly -> 422
    (reference anomaly.py:51-55)."""
    import shutil

    from sklearn.preprocessing import StandardScaler

    from gordo_amd import serializer

    src = os.path.join(model_collection_directory, "machine-1")
    plain = os.path.join(model_collection_directory, "plain-model")
    shutil.copytree(src, plain)
    scaler = StandardScaler()
    scaler.fit(np.random.default_rng(0).random((20, len(sensors))))
    serializer.dump(scaler, plain)

    client = flask_app.test_client()
    X = pd.DataFrame(
        np.random.default_rng(1).random((10, len(sensors))),
        columns=sensors,
        index=pd.date_range("2019-01-01", periods=10, freq="10min", tz="UTC"),
    )
    body = {"X": server_utils.dataframe_to_dict(X),
            "y": server_utils.dataframe_to_dict(X)}
    try:
        resp = client.post(
            f"/gordo/v0/{gordo_project}/plain-model/anomaly/prediction",
            json=body,
        )
        assert resp.status_code == 422
    finally:
        shutil.rmtree(plain)
        server_utils.load_model.cache_clear()


def test_frame_json_response_without_extension(monkeypatch):
    """Python-codec fallback when the C++ encoder is unavailable."""
    from flask import Flask

    from gordo_amd.server import utils as su

    monkeypatch.setattr(su, "_gordo_fastjson", None)
    df = pd.DataFrame(
        [[1.0, 2.0]],
        columns=pd.MultiIndex.from_tuples([("a", "x"), ("a", "y")]),
        index=["i"],
    )
    app = Flask("t2")
    with app.test_request_context():
        resp = su.frame_json_response({"extra": "1"}, df)
    body = json.loads(resp.get_data())
    assert body["data"]["a"] == {"x": {"i": 1.0}, "y": {"i": 2.0}}
    assert body["extra"] == "1"


@pytest.mark.timeout(120)
def test_run_server_prefork_workers(tmp_path):
    """Without gunicorn, --workers N serves through a prefork werkzeug
    pool: N worker processes share one listening socket (the
    reference's gunicorn master/worker model, server.py:240-304), all
    answer requests, and the pool restarts a killed worker."""
    import signal
    import socket
    import subprocess
    import sys
    import time
    import urllib.request

    import psutil

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    collection = tmp_path / "collection"
    collection.mkdir()
    env = dict(os.environ)
    env.update(
        MODEL_COLLECTION_DIR=str(collection),
        CUDA_VISIBLE_DEVICES="", HIP_VISIBLE_DEVICES="",
        ROCR_VISIBLE_DEVICES="",
    )
    proc = subprocess.Popen(
        [sys.executable, "-m", "gordo_amd", "run-server",
         "--host", "127.0.0.1", "--port", str(port), "--workers", "2"],
        env=env, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
    )
    try:
        url = f"http://127.0.0.1:{port}/healthcheck"
        deadline = time.time() + 60
        while True:
            try:
                with urllib.request.urlopen(url, timeout=2) as r:
                    assert r.status == 200
                    break
            except Exception:
                if time.time() > deadline:
                    raise
                time.sleep(0.3)
        parent = psutil.Process(proc.pid)
        kids = parent.children()
        assert len(kids) == 2, kids
        # requests keep flowing with both workers up
        for _ in range(4):
            with urllib.request.urlopen(url, timeout=5) as r:
                assert r.status == 200
        # supervision: kill one worker; the master respawns it
        kids[0].send_signal(signal.SIGKILL)
        deadline = time.time() + 30
        while True:
            alive = [c for c in parent.children() if c.is_running()]
            if len(alive) == 2 and kids[0].pid not in [c.pid for c in alive]:
                break
            assert time.time() < deadline, alive
            time.sleep(0.3)
        with urllib.request.urlopen(url, timeout=5) as r:
            assert r.status == 200
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=20)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=10)


def test_dataframe_from_dict_fast_lane_equivalence():
    """The serving-path fast decoder produces exactly what the
    reference-semantics path produces for every shape it accepts, and
    falls back for everything else."""
    import numpy as np
    import pandas as pd

    from gordo_amd.server import utils as su

    def legacy(data):
        # force the reference path by bypassing the fast lane
        import unittest.mock as mock

        with mock.patch.object(su, "_dataframe_from_dict_fast",
                               lambda d: None):
            return su.dataframe_from_dict(data)

    idx = pd.date_range("2020-01-01", periods=5, freq="10min", tz="UTC")
    frame = pd.DataFrame(
        np.random.random((5, 3)), index=idx, columns=["a", "b", "c"]
    )
    payloads = [
        su.dataframe_to_dict(frame),                       # tz-aware
        su.dataframe_to_dict(frame.tz_localize(None)),     # naive
        {"x": {"0": 1, "1": 2}, "y": {"0": 3, "1": 4}},    # int index+vals
        {"x": {"1": 0.5, "0": 1.5}},                       # unsorted
        # mixed dtypes per column (ints stay int64)
        {"i": {"2020-01-01": 1, "2020-01-02": 2},
         "f": {"2020-01-01": 0.5, "2020-01-02": 1.5}},
    ]
    for payload in payloads:
        fast = su.dataframe_from_dict(payload)
        ref = legacy(payload)
        pd.testing.assert_frame_equal(fast, ref)

    # shapes the fast lane must refuse (falls back, still correct)
    odd = {"x": {"0": 1.0}, "y": {"1": 2.0}}  # mismatched keys
    assert su._dataframe_from_dict_fast(odd) is None
    assert su.dataframe_from_dict(odd) is not None
    nested = {
        "top": {"sub": {"2019-01-01": 0.0, "2019-01-02": 1.0}}
    }  # 2-level (anomaly-style) dict
    assert su._dataframe_from_dict_fast(nested) is None
    assert su.dataframe_from_dict(nested).shape == (2, 1)


def test_decode_request_fast_equivalence():
    """The C++ strict-lane request decoder matches json.loads +
    dataframe_from_dict for everything it accepts, and refuses
    anything irregular."""
    import json as _json

    import numpy as np
    import pandas as pd
    import pytest as _pytest

    from gordo_amd.server import utils as su

    fj = su._gordo_fastjson
    if fj is None or not hasattr(fj, "decode_request"):
        _pytest.skip("fastjson not built")

    idx = pd.date_range("2020-01-01", periods=7, freq="10min", tz="UTC")
    X = pd.DataFrame(np.random.random((7, 4)),
                     index=idx, columns=["t 1", "t2", "t3", "t4"])
    X.iloc[0, 1] = np.nan
    for payload in (
        {"X": su.dataframe_to_dict(X)},
        {"X": su.dataframe_to_dict(X), "y": su.dataframe_to_dict(X)},
        {"X": su.dataframe_to_dict(X), "y": None},
        {"X": {"a": {"0": 1, "1": 2}}},  # int index + int values
    ):
        raw = _json.dumps(payload).encode()
        fast = su._decode_request_fast(raw)
        assert fast is not None, payload.keys()
        want_X = su.dataframe_from_dict(payload["X"])
        pd.testing.assert_frame_equal(fast["X"], want_X)
        if payload.get("y") is not None:
            pd.testing.assert_frame_equal(
                fast["y"], su.dataframe_from_dict(payload["y"])
            )
    # NaN round trip: stdlib emits NaN token, decoder must accept it
    raw = _json.dumps({"X": su.dataframe_to_dict(X)}).encode()
    assert b"NaN" in raw
    fast = su._decode_request_fast(raw)
    assert np.isnan(fast["X"].iloc[0, 1])

    # refused shapes -> None (stdlib path takes over)
    for bad in (
        b'{"X": {"a": {"k\\u0041": 1}}}',          # escaped key
        b'{"X": {"a": {"0": 1}}, "extra": 2}',     # unknown top key
        b'{"X": {"a": {"0": 1}, "b": {"1": 1}}}',  # key-set mismatch
        b'{"X": {"a": {"0": "s"}}}',               # non-number value
        b'{"X": []}',                              # wrong shape
        b'not json',
    ):
        assert su._decode_request_fast(bad) is None
