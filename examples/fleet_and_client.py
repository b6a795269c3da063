"""
Fleet example: build a mixed feedforward/LSTM fleet across all visible
GPUs (CPU fallback), serve it, and consume it with the Client.

Run: python examples/fleet_and_client.py
"""
import json
import os
import sys
import tempfile
import threading

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import dateutil.parser

from gordo_amd.client import Client
from gordo_amd.machine import Machine
from gordo_amd.parallel import PackedFleetBuilder
from gordo_amd.workflow import NormalizedConfig

TAGS = [f"sensor-{i}" for i in range(12)]


def machine_cfg(name, kind):
    model = {
        "gordo.machine.model.anomaly.diff.DiffBasedAnomalyDetector": {
            "require_thresholds": True,
            "base_estimator": {
                "sklearn.pipeline.Pipeline": {
                    "steps": [
                        "sklearn.preprocessing.MinMaxScaler",
                        {
                            (
                                "gordo.machine.model.models.KerasLSTMAutoEncoder"
                                if kind == "lstm"
                                else "gordo.machine.model.models.KerasAutoEncoder"
                            ): (
                                {"kind": "lstm_hourglass",
                                 "lookback_window": 24, "epochs": 2}
                                if kind == "lstm"
                                else {"kind": "feedforward_hourglass",
                                      "epochs": 3}
                            )
                        },
                    ]
                }
            },
        }
    }
    return {
        "name": name,
        "model": model,
        "dataset": {
            "type": "SineWaveDataset",
            "tag_list": TAGS,
            "train_start_date": "2019-01-01T00:00:00+00:00",
            "train_end_date": "2019-01-08T00:00:00+00:00",
        },
    }


config = {
    "machines": [
        machine_cfg("pump-01", "dense"),
        machine_cfg("pump-02", "dense"),
        machine_cfg("compressor-01", "lstm"),
    ]
}

with tempfile.TemporaryDirectory() as tmp:
    collection = os.path.join(tmp, "1")
    norm = NormalizedConfig(config, project_name="fleet-example")
    results = PackedFleetBuilder(norm.machines, output_dir=collection).build_all()
    for name, res in results:
        ok = not isinstance(res, BaseException)
        print(f"built {name}: " + ("OK" if ok else repr(res)))

    os.environ["MODEL_COLLECTION_DIR"] = collection
    from gordo_amd.server.server import build_app

    app = build_app()
    app.testing = True

    class FlaskSession:  # route client HTTP into the in-process app
        def __init__(self):
            self.client = app.test_client()
            self.lock = threading.Lock()

        def request(self, method, url, params=None, json=None, **kw):
            path = "/" + url.split("://", 1)[-1].split("/", 1)[1]
            with self.lock:
                return self.client.open(path, method=method, json=json,
                                        query_string=params)

    client = Client(project="fleet-example", session=FlaskSession())
    print("served models:", client.get_machine_names())
    start = dateutil.parser.isoparse("2019-01-01T00:00:00+00:00")
    end = dateutil.parser.isoparse("2019-01-02T00:00:00+00:00")
    for name, frame, errors in client.predict(start, end):
        conf = frame["total-anomaly-confidence"].to_numpy().max()
        print(f"{name}: {len(frame)} scored rows, errors={errors}, "
              f"max total-anomaly-confidence={float(conf):.3f}")
