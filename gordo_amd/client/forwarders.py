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
    """Influx forwarder over the plain 1.x HTTP write API (line
    protocol via ``requests`` — the ``influxdb`` client package is not
    installed in this image). Behavioral spec: gordo-client's
    ForwardPredictionsIntoInflux (reference
    tests/gordo/client/test_client.py; Argo client pods, template
    :1375): each top-level column family of the prediction frame
    becomes one measurement point stream tagged with the machine name.
    """

    def __init__(
        self,
        destination_influx_uri: Optional[str] = None,
        destination_influx_api_key: Optional[str] = None,
        destination_influx_recreate: bool = False,
        n_retries: int = 5,
        batch_size: int = 5000,
        timeout: float = 30.0,
    ):
        from ..core.data_providers import parse_influx_uri

        self.destination_influx_uri = destination_influx_uri
        self.destination_influx_api_key = destination_influx_api_key
        self.destination_influx_recreate = destination_influx_recreate
        self.n_retries = n_retries
        self.batch_size = batch_size
        self.timeout = timeout
        self.base_url, self.auth, self.database = parse_influx_uri(
            destination_influx_uri
        )

    @staticmethod
    def _escape(s: str) -> str:
        return (
            str(s).replace("\\", "\\\\").replace(" ", "\\ ")
            .replace(",", "\\,").replace("=", "\\=")
        )

    def _lines(self, predictions: pd.DataFrame, machine_name: str):
        """Line-protocol points: measurement per top-level column
        family, fields per sub-column, tagged machine=<name>."""
        cols = predictions.columns
        if not isinstance(cols, pd.MultiIndex):
            cols = pd.MultiIndex.from_tuples([(c, "value") for c in cols])
        index_ns = pd.DatetimeIndex(predictions.index).asi8
        values = predictions.to_numpy()
        by_family: dict = {}
        for j, (top, sub) in enumerate(cols):
            by_family.setdefault(str(top), []).append((str(sub), j))
        mtag = self._escape(machine_name)
        for family, members in by_family.items():
            meas = self._escape(family)
            for i, ts in enumerate(index_ns):
                fields = ",".join(
                    f"{self._escape(sub)}={float(values[i, j])!r}"
                    for sub, j in members
                    if values[i, j] == values[i, j]  # drop NaN fields
                )
                if fields:
                    yield f"{meas},machine={mtag} {fields} {ts}"

    def _post_batch(self, body: str):
        import time as _time

        import requests

        headers = {}
        if self.destination_influx_api_key:
            headers["Ocp-Apim-Subscription-Key"] = (
                self.destination_influx_api_key
            )
        last: Optional[Exception] = None
        for attempt in range(self.n_retries):
            try:
                resp = requests.post(
                    f"{self.base_url}/write",
                    params={"db": self.database, "precision": "ns"},
                    data=body.encode(),
                    headers=headers,
                    auth=self.auth,
                    timeout=self.timeout,
                )
                resp.raise_for_status()
                return
            except Exception as e:  # exponential backoff like the client
                last = e
                _time.sleep(min(2 ** attempt * 0.1, 5.0))
        raise RuntimeError(
            f"influx write failed after {self.n_retries} retries"
        ) from last

    def forward_predictions(self, predictions, machine_name, metadata=None):
        if self.destination_influx_recreate:
            self._recreate_database()
        batch: list = []
        n = 0
        for line in self._lines(predictions, machine_name):
            batch.append(line)
            if len(batch) >= self.batch_size:
                self._post_batch("\n".join(batch))
                n += len(batch)
                batch = []
        if batch:
            self._post_batch("\n".join(batch))
            n += len(batch)
        logger.info(
            "Forwarded %d influx points for %s -> %s",
            n, machine_name, self.base_url,
        )

    def _recreate_database(self):
        import requests

        for q in (
            f'DROP DATABASE "{self.database}"',
            f'CREATE DATABASE "{self.database}"',
        ):
            requests.post(
                f"{self.base_url}/query", params={"q": q},
                auth=self.auth, timeout=self.timeout,
            ).raise_for_status()
