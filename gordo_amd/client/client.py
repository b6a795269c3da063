"""
Client — programmatic access to a gordo server deployment.

Behavioral spec: the external ``gordo-client`` package as exercised by
the reference test-suite (tests/gordo/client/test_client.py:38-497):
``Client(project=..., host=..., port=..., scheme=..., parallelism=...,
metadata=...)`` with ``predict(start, end, targets)`` batched over date
ranges, ``get_metadata()``, ``download_model()``, revision handling and
retry/backoff on transient errors.

The HTTP transport is pluggable (``session=``): production uses a
``requests.Session``; the test-suite routes the same calls into an
in-process Flask test client (tests/test_client.py — the reference's
"mock ML-server mesh" pattern, conftest.py:333-422).
"""
from __future__ import annotations

import logging
import time
from concurrent.futures import ThreadPoolExecutor
from datetime import datetime
from typing import Any, Dict, List, Optional, Tuple

import pandas as pd

from ..server.utils import dataframe_from_dict, dataframe_to_dict

logger = logging.getLogger(__name__)


class HttpUnprocessableEntity(Exception):
    """422 from the server (e.g. model is not an anomaly detector)."""


class ResourceGone(Exception):
    """410 — requested revision no longer exists."""


class NotFound(Exception):
    """404 — no such model."""


class BadGordoRequest(Exception):
    """4xx the client considers non-retryable."""


class Client:
    def __init__(
        self,
        project: str,
        host: str = "localhost",
        port: int = 443,
        scheme: str = "https",
        metadata: Optional[dict] = None,
        parallelism: int = 10,
        n_retries: int = 5,
        use_parquet: bool = False,
        session: Optional[Any] = None,
        batch_size: int = 1000,
        revision: Optional[str] = None,
    ):
        self.project_name = project
        self.base_url = f"{scheme}://{host}:{port}"
        self.metadata = metadata if metadata is not None else {}
        self.parallelism = parallelism
        self.n_retries = n_retries
        self.use_parquet = use_parquet
        self.batch_size = batch_size
        self.revision = revision
        if session is None:
            import requests

            session = requests.Session()
        self.session = session

    # ---- plumbing ------------------------------------------------------
    def _url(self, path: str) -> str:
        return f"{self.base_url}/gordo/v0/{self.project_name}/{path}"

    def _request(self, method: str, url: str, **kwargs):
        last_exc: Optional[Exception] = None
        for attempt in range(self.n_retries):
            try:
                resp = self.session.request(method, url, **kwargs)
            except Exception as e:  # transport error: retry with backoff
                last_exc = e
                time.sleep(min(2 ** attempt * 0.1, 5.0))
                continue
            if resp.status_code == 422:
                raise HttpUnprocessableEntity(url)
            if resp.status_code == 410:
                raise ResourceGone(url)
            if resp.status_code == 404:
                raise NotFound(url)
            if 400 <= resp.status_code < 500:
                raise BadGordoRequest(f"{url} -> {resp.status_code}")
            if resp.status_code >= 500:
                last_exc = RuntimeError(f"{url} -> {resp.status_code}")
                time.sleep(min(2 ** attempt * 0.1, 5.0))
                continue
            return resp
        raise last_exc if last_exc else RuntimeError(f"request failed: {url}")

    @staticmethod
    def _json(resp) -> dict:
        if hasattr(resp, "json"):
            body = resp.json
            return body() if callable(body) else body
        raise ValueError("response has no json")

    # ---- API -----------------------------------------------------------
    def get_revisions(self) -> dict:
        return self._json(self._request("GET", self._url("revisions")))

    def get_available_machines(self, revision: Optional[str] = None) -> dict:
        params = {"revision": revision or self.revision}
        params = {k: v for k, v in params.items() if v}
        return self._json(
            self._request("GET", self._url("models"), params=params)
        )

    def get_machine_names(self, revision: Optional[str] = None) -> List[str]:
        return self.get_available_machines(revision).get("models", [])

    def get_metadata(
        self, revision: Optional[str] = None, targets: Optional[List[str]] = None
    ) -> Dict[str, dict]:
        names = targets or self.get_machine_names(revision)
        params = {"revision": revision or self.revision}
        params = {k: v for k, v in params.items() if v}

        def fetch(name):
            resp = self._request(
                "GET", self._url(f"{name}/metadata"), params=params
            )
            return name, self._json(resp)["metadata"]

        with ThreadPoolExecutor(max_workers=self.parallelism) as ex:
            return dict(ex.map(fetch, names))

    def download_model(
        self, revision: Optional[str] = None, targets: Optional[List[str]] = None
    ) -> Dict[str, Any]:
        from .. import serializer

        names = targets or self.get_machine_names(revision)
        out = {}
        for name in names:
            resp = self._request("GET", self._url(f"{name}/download-model"))
            data = resp.data if hasattr(resp, "data") else resp.content
            out[name] = serializer.loads(data)
        return out

    def predict(
        self,
        start: datetime,
        end: datetime,
        targets: Optional[List[str]] = None,
        revision: Optional[str] = None,
    ) -> List[Tuple[str, pd.DataFrame, List[str]]]:
        """
        Anomaly predictions for each target machine over [start, end),
        fetched through each machine's own dataset config, batched over
        sub-ranges and machines in parallel.

        Returns [(machine_name, anomaly_frame, error_messages)].
        """
        metadata = self.get_metadata(revision, targets)
        with ThreadPoolExecutor(max_workers=self.parallelism) as ex:
            return list(
                ex.map(
                    lambda item: self._predict_single(
                        item[0], item[1], start, end, revision
                    ),
                    metadata.items(),
                )
            )

    def _predict_single(self, name, machine_meta, start, end, revision):
        errors: List[str] = []
        frames: List[pd.DataFrame] = []
        try:
            from ..core.base import GordoBaseDataset

            dataset_cfg = dict(machine_meta["dataset"])
            dataset_cfg["train_start_date"] = start
            dataset_cfg["train_end_date"] = end
            dataset = GordoBaseDataset.from_dict(dataset_cfg)
            X, y = dataset.get_data()
            for s in range(0, len(X), self.batch_size):
                Xb, yb = X.iloc[s : s + self.batch_size], y.iloc[s : s + self.batch_size]
                payload = {
                    "X": dataframe_to_dict(Xb),
                    "y": dataframe_to_dict(yb),
                }
                params = {"revision": revision or self.revision}
                params = {k: v for k, v in params.items() if v}
                resp = self._request(
                    "POST",
                    self._url(f"{name}/anomaly/prediction"),
                    json=payload,
                    params=params,
                )
                frames.append(dataframe_from_dict(self._json(resp)["data"]))
        except HttpUnprocessableEntity:
            errors.append(f"Target {name} does not support anomaly predictions")
        except Exception as e:
            errors.append(f"Failed to predict target {name}: {e}")
        frame = pd.concat(frames) if frames else pd.DataFrame()
        return name, frame, errors
