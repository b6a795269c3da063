"""
Minimal end-to-end example: define two machines in YAML, build them
in-process, dump to a model collection dir, serve with Flask, and
query an anomaly prediction.

Run: python examples/minimal_build_and_serve.py
"""
import json
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd

from gordo_amd import serializer
from gordo_amd.builder import local_build
from gordo_amd.server.utils import dataframe_to_dict

CONFIG = """
machines:
  - name: example-machine
    dataset: |
      type: SineWaveDataset
      tags: [pressure, temperature, flow]
      train_start_date: '2019-01-01T00:00:00+00:00'
      train_end_date: '2019-01-08T00:00:00+00:00'
    model: |
      gordo.machine.model.anomaly.diff.DiffBasedAnomalyDetector:
        require_thresholds: true
        base_estimator:
          sklearn.pipeline.Pipeline:
            steps:
            - sklearn.preprocessing.MinMaxScaler
            - gordo.machine.model.models.KerasAutoEncoder:
                kind: feedforward_hourglass
                epochs: 5
"""

with tempfile.TemporaryDirectory() as tmp:
    collection = os.path.join(tmp, "1")
    for model, machine in local_build(CONFIG, project_name="example"):
        out = os.path.join(collection, machine.name)
        serializer.dump(model, out, metadata=json.loads(machine.to_json()),
                        info={})
        print(f"built {machine.name} -> {out}")

    os.environ["MODEL_COLLECTION_DIR"] = collection
    from gordo_amd.server.server import build_app

    client = build_app().test_client()
    X = pd.DataFrame(np.random.rand(20, 3),
                     columns=["pressure", "temperature", "flow"])
    resp = client.post(
        "/gordo/v0/example/example-machine/anomaly/prediction",
        json={"X": dataframe_to_dict(X), "y": dataframe_to_dict(X)},
    )
    assert resp.status_code == 200
    body = resp.json
    print("anomaly columns:", sorted(body["data"].keys()))
    print("total-anomaly-confidence[0]:",
          list(body["data"]["total-anomaly-confidence"][""].values())[0])
