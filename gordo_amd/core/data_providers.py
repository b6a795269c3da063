"""
Data providers — sources of raw per-tag timeseries.

Spec: gordo_core data providers as used by the reference
(``RandomDataProvider`` in tests/conftest.py; provider configured via
``dataset.data_provider.type``). Providers here are synthetic (there is
no external data store in this environment): random walks and sine
waves, deterministic per (tag, seed).
"""
from __future__ import annotations

import hashlib
from typing import Iterable, Optional, Dict, Any

import numpy as np
import pandas as pd

from .sensor_tag import SensorTag
from .import_utils import import_location


class DataProvider:
    """Base data provider: serves one pd.Series per SensorTag."""

    def __init__(self, **kwargs):
        self._params = dict(kwargs)

    @classmethod
    def from_dict(cls, config: Dict[str, Any]) -> "DataProvider":
        config = dict(config or {})
        kind = config.pop("type", None)
        if kind is None:
            return RandomDataProvider(**config)
        if "." in kind:
            provider_cls = import_location(kind)
        else:
            provider_cls = _PROVIDER_REGISTRY.get(kind)
            if provider_cls is None:
                from .exceptions import NoSuitableDataProviderError

                raise NoSuitableDataProviderError(
                    f"No data provider named {kind!r}"
                )
        return provider_cls(**config)

    def to_dict(self) -> Dict[str, Any]:
        d = dict(self._params)
        d["type"] = type(self).__name__
        return d

    def can_handle_tag(self, tag: SensorTag) -> bool:
        return True

    def load_series(
        self,
        train_start_date: pd.Timestamp,
        train_end_date: pd.Timestamp,
        tag_list: Iterable[SensorTag],
        resolution: str = "10T",
    ) -> Iterable[pd.Series]:
        raise NotImplementedError()

    def load_frame(
        self,
        train_start_date: pd.Timestamp,
        train_end_date: pd.Timestamp,
        tag_list: Iterable[SensorTag],
        resolution: str = "10T",
    ) -> Optional[pd.DataFrame]:
        """Whole-frame fast path: all tags in ONE aligned DataFrame
        (None -> caller falls back to per-series load_series + join).
        Synthetic providers emit naturally aligned data, so the
        per-series resample/concat of the general path is pure
        overhead (~30%% of a build step's data phase at bench shape).
        Values must be identical to load_series."""
        return None

    def __repr__(self):
        return f"{type(self).__name__}({self._params!r})"


def _tag_seed(tag: SensorTag, salt: int) -> int:
    h = hashlib.sha256(f"{tag.name}:{salt}".encode()).digest()
    return int.from_bytes(h[:4], "little")


def _time_index(start, end, resolution) -> pd.DatetimeIndex:
    return pd.date_range(start=start, end=end, freq=resolution, inclusive="left")


class RandomDataProvider(DataProvider):
    """Deterministic random-walk series per tag (test/synthetic data)."""

    def __init__(self, min_size: int = 100, max_size: int = 300, **kwargs):
        super().__init__(min_size=min_size, max_size=max_size, **kwargs)
        self.min_size = min_size
        self.max_size = max_size

    def load_series(self, train_start_date, train_end_date, tag_list, resolution="10T"):
        index = _time_index(train_start_date, train_end_date, resolution)
        for tag in tag_list:
            rng = np.random.default_rng(_tag_seed(tag, 0))
            values = rng.standard_normal(len(index)).cumsum() * 0.1 + rng.uniform(
                -1.0, 1.0
            )
            yield pd.Series(values, index=index, name=tag.name)


class SineWaveDataProvider(DataProvider):
    """
    Synthetic sine-wave tags (the benchmark data source — BASELINE.md:
    "synthetic sine-wave tag data"). Each tag gets a deterministic
    frequency/phase/noise from its name.
    """

    def __init__(self, noise: float = 0.02, **kwargs):
        super().__init__(noise=noise, **kwargs)
        self.noise = noise

    def _tag_values(self, tag, t):
        rng = np.random.default_rng(_tag_seed(tag, 1))
        freq = rng.uniform(0.002, 0.05)
        phase = rng.uniform(0, 2 * np.pi)
        amp = rng.uniform(0.5, 2.0)
        offset = rng.uniform(-1.0, 1.0)
        return (
            amp * np.sin(2 * np.pi * freq * t + phase)
            + offset
            + rng.standard_normal(len(t)) * self.noise
        )

    def load_series(self, train_start_date, train_end_date, tag_list, resolution="10T"):
        index = _time_index(train_start_date, train_end_date, resolution)
        t = np.arange(len(index), dtype=np.float64)
        for tag in tag_list:
            yield pd.Series(self._tag_values(tag, t), index=index, name=tag.name)

    def load_frame(self, train_start_date, train_end_date, tag_list, resolution="10T"):
        index = _time_index(train_start_date, train_end_date, resolution)
        t = np.arange(len(index), dtype=np.float64)
        tags = list(tag_list)
        out = np.empty((len(index), len(tags)), dtype=np.float64)
        for j, tag in enumerate(tags):
            out[:, j] = self._tag_values(tag, t)
        return pd.DataFrame(out, index=index, columns=[tg.name for tg in tags])


class InfluxDataProvider(DataProvider):
    """
    InfluxDB-backed provider over the plain 1.x HTTP API (the
    ``influxdb`` client package is not installed in this image, so this
    speaks ``GET /query`` JSON directly via ``requests`` — behavioral
    spec: gordo-core's InfluxDataProvider: one series per tag from
    ``SELECT "<value_name>" FROM "<measurement>" WHERE tag = ...``).

    ``uri`` accepts ``[user:pass@]host:port/dbname`` (the gordo
    convention) or a full ``http://host:port`` plus ``database=``.
    """

    def __init__(
        self,
        uri: Optional[str] = None,
        api_key: Optional[str] = None,
        api_key_header: str = "Ocp-Apim-Subscription-Key",
        database: Optional[str] = None,
        measurement: str = "resampled",
        value_name: str = "Value",
        timeout: float = 30.0,
        **kwargs,
    ):
        super().__init__(
            uri=uri, api_key=api_key, api_key_header=api_key_header,
            database=database, measurement=measurement,
            value_name=value_name, **kwargs,
        )
        self.base_url, self.auth, db = parse_influx_uri(uri)
        self.database = database or db
        self.api_key = api_key
        self.api_key_header = api_key_header
        self.measurement = measurement
        self.value_name = value_name
        self.timeout = timeout

    def _query(self, q: str):
        import requests

        headers = {}
        if self.api_key:
            headers[self.api_key_header] = self.api_key
        resp = requests.get(
            f"{self.base_url}/query",
            params={"db": self.database, "q": q, "epoch": "ns"},
            headers=headers,
            auth=self.auth,
            timeout=self.timeout,
        )
        resp.raise_for_status()
        payload = resp.json()
        results = payload.get("results", [])
        if results and "error" in results[0]:
            raise RuntimeError(f"influx query error: {results[0]['error']}")
        return results

    def load_series(self, train_start_date, train_end_date, tag_list,
                    resolution="10T"):
        for tag in tag_list:
            q = (
                f'SELECT "{self.value_name}" FROM "{self.measurement}" '
                f"WHERE (\"tag\" = '{tag.name}') "
                f"AND time >= '{pd.Timestamp(train_start_date).isoformat()}' "
                f"AND time <= '{pd.Timestamp(train_end_date).isoformat()}'"
            )
            results = self._query(q)
            series_list = (results[0] or {}).get("series") if results else None
            if not series_list:
                yield pd.Series(dtype="float64", name=tag.name)
                continue
            cols = series_list[0]["columns"]
            vals = series_list[0]["values"]
            ti = cols.index("time")
            vi = cols.index(self.value_name) if self.value_name in cols else 1
            index = pd.to_datetime(
                [row[ti] for row in vals], utc=True, unit="ns"
            )
            yield pd.Series(
                [row[vi] for row in vals], index=index, name=tag.name,
                dtype="float64",
            )


def parse_influx_uri(uri: Optional[str]):
    """``[user:pass@]host[:port][/db]`` or full http(s) URL ->
    (base_url, auth-tuple-or-None, db-or-None)."""
    if not uri:
        return "http://localhost:8086", None, None
    auth = None
    rest = uri
    scheme = "http"
    if "://" in rest:
        scheme, rest = rest.split("://", 1)
    if "@" in rest:
        creds, rest = rest.rsplit("@", 1)
        user, _, pw = creds.partition(":")
        auth = (user, pw)
    db = None
    if "/" in rest:
        rest, db = rest.split("/", 1)
    return f"{scheme}://{rest}", auth, (db or None)


_PROVIDER_REGISTRY: Dict[str, type] = {
    "RandomDataProvider": RandomDataProvider,
    "SineWaveDataProvider": SineWaveDataProvider,
    "InfluxDataProvider": InfluxDataProvider,
    "DataLakeProvider": RandomDataProvider,  # alias: no data lake here
}


def load_data_provider(config: Optional[Dict[str, Any]]) -> DataProvider:
    if config is None:
        return RandomDataProvider()
    if isinstance(config, DataProvider):
        return config
    return DataProvider.from_dict(config)
