"""Client tests against the in-process Flask server (the reference's
"mock ML-server mesh" pattern: every HTTP call routes into the test
client — tests/conftest.py:333-422 there; a transport shim here)."""
import threading

import dateutil.parser
import pandas as pd
import pytest

from gordo_amd.client import Client, ForwardPredictionsToDisk


class FlaskSession:
    """requests.Session-compatible shim over a Flask test client."""

    def __init__(self, flask_app):
        self.client = flask_app.test_client()
        self.lock = threading.Lock()
        self.fail_next = 0  # fault injection: 500s before succeeding

    def request(self, method, url, params=None, json=None, **kwargs):
        path = url.split("://", 1)[-1].split("/", 1)[1]
        with self.lock:
            if self.fail_next > 0:
                self.fail_next -= 1
                return FakeResponse(500)
            resp = self.client.open(
                "/" + path, method=method, json=json, query_string=params
            )
        return resp


class FakeResponse:
    def __init__(self, status_code):
        self.status_code = status_code


@pytest.fixture
def client(flask_app, gordo_project):
    return Client(
        project=gordo_project,
        host="server",
        port=80,
        scheme="http",
        session=FlaskSession(flask_app),
        parallelism=2,
        n_retries=3,
    )


def test_get_revisions(client, gordo_revision):
    revs = client.get_revisions()
    assert revs["latest"] == gordo_revision


def test_machine_names(client, gordo_name, second_gordo_name):
    names = client.get_machine_names()
    assert {gordo_name, second_gordo_name} <= set(names)


def test_get_metadata(client, gordo_name):
    meta = client.get_metadata()
    assert gordo_name in meta
    assert meta[gordo_name]["name"] == gordo_name


def test_download_model(client, gordo_name):
    models = client.download_model(targets=[gordo_name])
    assert hasattr(models[gordo_name], "anomaly")


def test_predict(client, gordo_name):
    start = dateutil.parser.isoparse("2019-01-01T00:00:00+00:00")
    end = dateutil.parser.isoparse("2019-01-02T00:00:00+00:00")
    results = client.predict(start, end, targets=[gordo_name])
    assert len(results) == 1
    name, frame, errors = results[0]
    assert name == gordo_name
    assert errors == []
    assert len(frame) > 0
    assert "total-anomaly-scaled" in {c[0] for c in frame.columns}


def test_predict_batched(client, gordo_name):
    client.batch_size = 50  # force multiple POST batches
    start = dateutil.parser.isoparse("2019-01-01T00:00:00+00:00")
    end = dateutil.parser.isoparse("2019-01-02T00:00:00+00:00")
    name, frame, errors = client.predict(start, end, targets=[gordo_name])[0]
    assert errors == []
    assert len(frame) == 144  # one day at 10min resolution


def test_retry_on_5xx(client, gordo_name):
    client.session.fail_next = 2  # two 500s, then success
    names = client.get_machine_names()
    assert gordo_name in names


def test_forward_to_disk(tmp_path, client, gordo_name):
    start = dateutil.parser.isoparse("2019-01-01T00:00:00+00:00")
    end = dateutil.parser.isoparse("2019-01-02T00:00:00+00:00")
    name, frame, errors = client.predict(start, end, targets=[gordo_name])[0]
    fwd = ForwardPredictionsToDisk(str(tmp_path))
    fwd.forward_predictions(frame, name)
    stored = pd.read_parquet(tmp_path / f"{name}.parquet")
    assert len(stored) == len(frame)


def test_status_code_exception_mapping(client, gordo_name):
    """4xx statuses map to typed exceptions (reference
    tests/gordo/client/test_client.py::test__handle_response_errors)."""
    from gordo_amd.client.client import (
        BadGordoRequest,
        HttpUnprocessableEntity,
        NotFound,
        ResourceGone,
    )

    with pytest.raises(HttpUnprocessableEntity):
        client._request("GET", client._url("bad_name!/metadata"))
    with pytest.raises(NotFound):
        client._request("GET", client._url("no-such-model/metadata"))
    with pytest.raises(ResourceGone):
        client._request(
            "GET", client._url("models"), params={"revision": "123"}
        )
    with pytest.raises(BadGordoRequest):
        # POST prediction without an X payload → 400
        client._request(
            "POST", client._url(f"{gordo_name}/prediction"), json={}
        )


def test_unknown_revision_raises(flask_app, gordo_project):
    from gordo_amd.client.client import ResourceGone

    bad = Client(
        project=gordo_project,
        host="server",
        port=80,
        scheme="http",
        session=FlaskSession(flask_app),
        revision="123",
    )
    with pytest.raises(ResourceGone):
        bad.get_machine_names()


def test_metadata_specific_targets(client, gordo_name, second_gordo_name):
    md = client.get_metadata(targets=[gordo_name])
    assert set(md) == {gordo_name}
    assert md[gordo_name]["name"] == gordo_name


@pytest.fixture
def cli_env(flask_app, monkeypatch, gordo_project):
    """Route `gordo client` CLI commands through the in-process server."""
    from gordo_amd.cli import client as cli_client

    def make(ctx_obj):
        return Client(
            project=ctx_obj["project"],
            host="server",
            port=80,
            scheme="http",
            parallelism=ctx_obj.get("parallelism", 2),
            metadata=ctx_obj.get("metadata"),
            session=FlaskSession(flask_app),
        )

    monkeypatch.setattr(cli_client, "make_client", make)
    return gordo_project


@pytest.mark.parametrize(
    "args",
    [
        ["--help"],
        ["--project", "p", "predict", "--help"],
        ["--project", "p", "metadata", "--help"],
        ["--project", "p", "download-model", "--help"],
    ],
)
def test_client_cli_exists(args):
    from click.testing import CliRunner

    from gordo_amd.cli.cli import gordo

    out = CliRunner().invoke(gordo, ["client"] + args)
    assert out.exit_code == 0, out.output


def test_client_cli_metadata(cli_env, gordo_name, tmp_path):
    import json as _json

    from click.testing import CliRunner

    from gordo_amd.cli.cli import gordo

    out = CliRunner().invoke(
        gordo,
        ["client", "--project", cli_env, "metadata", "--target", gordo_name],
    )
    assert out.exit_code == 0, out.output
    assert gordo_name in out.output

    out_file = tmp_path / "metadata.json"
    out = CliRunner().invoke(
        gordo,
        ["client", "--project", cli_env, "metadata",
         "--output-file", str(out_file), "--target", gordo_name],
    )
    assert out.exit_code == 0, out.output
    assert gordo_name in _json.loads(out_file.read_text())


def test_client_cli_download_model(cli_env, gordo_name, tmp_path):
    from click.testing import CliRunner
    from sklearn.base import BaseEstimator

    from gordo_amd import serializer
    from gordo_amd.cli.cli import gordo

    out = CliRunner().invoke(
        gordo,
        ["client", "--project", cli_env, "download-model", str(tmp_path),
         "--target", gordo_name],
    )
    assert out.exit_code == 0, out.output
    model = serializer.load(str(tmp_path / gordo_name))
    assert isinstance(model, BaseEstimator)


def test_client_cli_predict(cli_env, gordo_name, tmp_path):
    from click.testing import CliRunner

    from gordo_amd.cli.cli import gordo

    fwd_dir = tmp_path / "fwd"
    out = CliRunner().invoke(
        gordo,
        ["client", "--project", cli_env, "predict",
         "2019-01-01T00:00:00Z", "2019-01-01T06:00:00Z",
         "--target", gordo_name, "--output-dir", str(tmp_path),
         "--forward-to-disk", str(fwd_dir)],
    )
    assert out.exit_code == 0, out.output
    saved = list(tmp_path.glob("*.csv.gz"))
    assert len(saved) == 1 and gordo_name in saved[0].name
    assert (fwd_dir / f"{gordo_name}.parquet").is_file()


def test_client_predict_all_targets(client, gordo_name, second_gordo_name):
    """predict() without explicit targets covers every served model."""
    import dateutil.parser

    start = dateutil.parser.isoparse("2019-01-01T00:00:00Z")
    end = dateutil.parser.isoparse("2019-01-01T06:00:00Z")
    results = client.predict(start, end)
    names = {name for name, _, _ in results}
    assert {gordo_name, second_gordo_name} <= names
    for name, frame, errors in results:
        assert not errors, (name, errors)
        assert len(frame) > 0


def test_client_cli_metadata_all_targets(cli_env, gordo_name,
                                         second_gordo_name):
    from click.testing import CliRunner

    from gordo_amd.cli.cli import gordo

    out = CliRunner().invoke(
        gordo, ["client", "--project", cli_env, "metadata"]
    )
    assert out.exit_code == 0, out.output
    assert gordo_name in out.output and second_gordo_name in out.output
