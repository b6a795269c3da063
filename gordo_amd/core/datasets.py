"""
TimeSeriesDataset / RandomDataset — fetch, resample, align.

Behavioral spec: ``get_data()`` fetches one series per
tag from the data provider, resamples each to ``resolution``
(aggregating), inner-joins on timestamps, applies ``row_filter`` and
returns ``(X, y)`` where X = tag_list columns, y = target_tag_list
columns. Accepted config keys mirror the reference's TimeSeriesDataset
(gordo/machine/validators.py:59-69, gordo/server/properties.py:62-69,
gordo/reporters/mlflow.py:219-225).
"""
from __future__ import annotations

import logging
from typing import Any, Dict, List, Optional, Tuple, Union

import dateutil.parser
import numpy as np
import pandas as pd

from .base import GordoBaseDataset
from .data_providers import DataProvider, RandomDataProvider, load_data_provider
from .exceptions import ConfigException, InsufficientDataError
from .sensor_tag import normalize_sensor_tag

logger = logging.getLogger(__name__)


def _parse_datetime(value) -> pd.Timestamp:
    if isinstance(value, pd.Timestamp):
        ts = value
    elif hasattr(value, "isoformat"):
        ts = pd.Timestamp(value)
    elif isinstance(value, str):
        ts = pd.Timestamp(dateutil.parser.isoparse(value))
    else:
        raise ConfigException(f"Cannot parse datetime from {value!r}")
    if ts.tzinfo is None:
        raise ConfigException(
            f"Datetime {value!r} lacks timezone information (tz is required)"
        )
    return ts


class TimeSeriesDataset(GordoBaseDataset):
    def __init__(
        self,
        train_start_date,
        train_end_date,
        tag_list: Optional[List] = None,
        tags: Optional[List] = None,
        target_tag_list: Optional[List] = None,
        resolution: str = "10min",
        data_provider: Optional[Union[Dict[str, Any], DataProvider]] = None,
        row_filter: Union[str, list] = "",
        row_filter_buffer_size: int = 0,
        aggregation_methods: str = "mean",
        n_samples_threshold: int = 0,
        filter_periods: Optional[dict] = None,
        asset: Optional[str] = None,
        default_asset: Optional[str] = None,
        default_tag: Optional[dict] = None,
        **kwargs,
    ):
        if tag_list is None and tags is None:
            raise ConfigException("TimeSeriesDataset requires 'tag_list' (or 'tags')")
        raw_tags = tag_list if tag_list is not None else tags
        self.asset = asset or default_asset
        self.tag_list = [normalize_sensor_tag(t, self.asset) for t in raw_tags]
        self.target_tag_list = (
            [normalize_sensor_tag(t, self.asset) for t in target_tag_list]
            if target_tag_list
            else list(self.tag_list)
        )
        self.train_start_date = _parse_datetime(train_start_date)
        self.train_end_date = _parse_datetime(train_end_date)
        if self.train_start_date >= self.train_end_date:
            raise ConfigException(
                f"train_start_date ({self.train_start_date}) must be before "
                f"train_end_date ({self.train_end_date})"
            )
        # "10T" (reference-era pandas) and "10min" both accepted
        self.resolution = str(resolution).replace("T", "min") if str(
            resolution
        ).endswith("T") else str(resolution)
        self.data_provider = load_data_provider(data_provider)
        self.row_filter = row_filter
        self.row_filter_buffer_size = int(row_filter_buffer_size)
        self.aggregation_methods = aggregation_methods
        self.n_samples_threshold = int(n_samples_threshold)
        self.filter_periods = filter_periods
        self.default_tag = default_tag
        self._extra = dict(kwargs)
        self._metadata: Dict[str, Any] = {}

        self._params = {
            "train_start_date": self.train_start_date.isoformat(),
            "train_end_date": self.train_end_date.isoformat(),
            "tag_list": [t.to_json() for t in self.tag_list],
            "target_tag_list": [t.to_json() for t in self.target_tag_list],
            "resolution": self.resolution,
            "data_provider": self.data_provider.to_dict(),
            "row_filter": self.row_filter,
            "row_filter_buffer_size": self.row_filter_buffer_size,
            "aggregation_methods": self.aggregation_methods,
            "n_samples_threshold": self.n_samples_threshold,
        }
        if self.asset:
            self._params["asset"] = self.asset
        if self.filter_periods:
            self._params["filter_periods"] = self.filter_periods
        if self.default_tag:
            self._params["default_tag"] = self.default_tag

    def get_data(self) -> Tuple[pd.DataFrame, pd.DataFrame]:
        all_tags = {t.name: t for t in self.tag_list}
        for t in self.target_tag_list:
            all_tags.setdefault(t.name, t)

        frame = self.data_provider.load_frame(
            self.train_start_date,
            self.train_end_date,
            list(all_tags.values()),
            resolution=self.resolution,
        )
        if frame is not None:
            # aligned single-frame fast path (values identical to the
            # per-series join; provider contract in DataProvider.load_frame)
            frame = frame[~np.isnan(frame.values).any(axis=1)]
        else:
            series_list = []
            for series in self.data_provider.load_series(
                self.train_start_date,
                self.train_end_date,
                list(all_tags.values()),
                resolution=self.resolution,
            ):
                series_list.append(self._resample(series))

            if not series_list:
                raise InsufficientDataError("Data provider returned no series")

            frame = pd.concat(series_list, axis=1, join="inner")
            frame = frame.dropna(how="any")

        if self.row_filter:
            frame = self._apply_row_filter(frame)

        if len(frame) <= self.n_samples_threshold:
            raise InsufficientDataError(
                f"Dataset has {len(frame)} rows; threshold is "
                f"{self.n_samples_threshold}"
            )

        X = frame[[t.name for t in self.tag_list]]
        y = frame[[t.name for t in self.target_tag_list]]
        self._metadata = {
            "tag_loading_metadata": {
                "tags": {name: tag.to_json() for name, tag in all_tags.items()},
                "rows": len(frame),
            },
            "dataset_meta": {
                "row_count": len(frame),
                "resolution": self.resolution,
            },
        }
        return X, y

    def _resample(self, series: pd.Series) -> pd.Series:
        # fast path: the series is already exactly on the target grid
        # (synthetic providers generate aligned data) — resampling would
        # be an expensive no-op per tag.
        freq = getattr(series.index, "freq", None)
        if freq is not None and freq == pd.tseries.frequencies.to_offset(
            self.resolution
        ):
            return series
        agg = self.aggregation_methods
        resampled = series.resample(self.resolution)
        if isinstance(agg, str):
            return getattr(resampled, agg)()
        return resampled.agg(agg)

    def _apply_row_filter(self, frame: pd.DataFrame) -> pd.DataFrame:
        filters = (
            [self.row_filter] if isinstance(self.row_filter, str) else self.row_filter
        )
        mask = pd.Series(True, index=frame.index)
        for f in filters:
            if not f:
                continue
            mask &= frame.eval(f, engine="python")
        if self.row_filter_buffer_size > 0:
            # drop rows within buffer_size of a filtered-out row
            bad = ~mask
            buffered = (
                bad.rolling(2 * self.row_filter_buffer_size + 1, center=True, min_periods=1)
                .max()
                .astype(bool)
            )
            mask = ~buffered
        return frame[mask]

    def get_metadata(self) -> Dict[str, Any]:
        meta = dict(self._metadata)
        meta.update(
            {
                "train_start_date": self.train_start_date.isoformat(),
                "train_end_date": self.train_end_date.isoformat(),
                "resolution": self.resolution,
                "tag_list": [t.to_json() for t in self.tag_list],
                "target_tag_list": [t.to_json() for t in self.target_tag_list],
            }
        )
        return meta


class RandomDataset(TimeSeriesDataset):
    """TimeSeriesDataset defaulting to the RandomDataProvider
    (reference: used at gordo/builder/build_model.py:71 and throughout
    the test-suite)."""

    def __init__(self, train_start_date, train_end_date, tag_list=None, **kwargs):
        kwargs.setdefault("data_provider", RandomDataProvider())
        super().__init__(
            train_start_date=train_start_date,
            train_end_date=train_end_date,
            tag_list=tag_list,
            **kwargs,
        )
        self._params["type"] = "RandomDataset"

    def to_dict(self):
        d = super().to_dict()
        d["type"] = "RandomDataset"
        return d


class SineWaveDataset(TimeSeriesDataset):
    """TimeSeriesDataset over synthetic sine tags — the benchmark data
    source."""

    def __init__(self, train_start_date, train_end_date, tag_list=None, **kwargs):
        from .data_providers import SineWaveDataProvider

        kwargs.setdefault("data_provider", SineWaveDataProvider())
        super().__init__(
            train_start_date=train_start_date,
            train_end_date=train_end_date,
            tag_list=tag_list,
            **kwargs,
        )
        self._params["type"] = "SineWaveDataset"
