"""
Prediction forwarders — push client prediction frames into a
timeseries store.

Spec: gordo-client's ``ForwardPredictionsIntoInflux`` as exercised by
the reference (tests/gordo/client/test_client.py; Argo client pods
backfill predictions into InfluxDB, template :1375). InfluxDB is not
reachable in this environment, so the Influx forwarder degrades to a
clear error at forward time while the base class and a CSV forwarder
(useful for local runs) are fully functional.
"""
from __future__ import annotations

import abc
import logging
import os
from typing import Optional

import pandas as pd

logger = logging.getLogger(__name__)


class PredictionForwarder(abc.ABC):
    @abc.abstractmethod
    def forward_predictions(
        self,
        predictions: pd.DataFrame,
        machine_name: str,
        metadata: Optional[dict] = None,
    ):
        ...


class ForwardPredictionsToDisk(PredictionForwarder):
    """Write each machine's prediction frame as parquet under a dir."""

    def __init__(self, destination_dir: str):
        self.destination_dir = destination_dir

    def forward_predictions(self, predictions, machine_name, metadata=None):
        os.makedirs(self.destination_dir, exist_ok=True)
        path = os.path.join(self.destination_dir, f"{machine_name}.parquet")
        flat = predictions.copy()
        if isinstance(flat.columns, pd.MultiIndex):
            flat.columns = ["::".join(map(str, c)).rstrip(":") for c in flat.columns]
        flat.to_parquet(path)
        logger.info("Forwarded %d rows for %s -> %s", len(flat), machine_name, path)


class ForwardPredictionsIntoInflux(PredictionForwarder):
    """Influx forwarder (API-compatible constructor; requires a
    reachable InfluxDB + the influxdb package at forward time)."""

    def __init__(
        self,
        destination_influx_uri: Optional[str] = None,
        destination_influx_api_key: Optional[str] = None,
        destination_influx_recreate: bool = False,
        n_retries: int = 5,
    ):
        self.destination_influx_uri = destination_influx_uri
        self.destination_influx_api_key = destination_influx_api_key
        self.destination_influx_recreate = destination_influx_recreate
        self.n_retries = n_retries

    def forward_predictions(self, predictions, machine_name, metadata=None):
        try:
            import influxdb  # noqa: F401
        except ImportError as e:
            raise RuntimeError(
                "influxdb client library is not installed in this environment"
            ) from e
        raise RuntimeError(
            "No reachable InfluxDB in this environment; use "
            "ForwardPredictionsToDisk for local runs"
        )
