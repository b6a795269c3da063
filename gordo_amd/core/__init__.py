"""
In-repo data layer (the reference consumed this surface from the
external ``gordo_core`` package — SURVEY.md §2.2).
"""
from .base import GordoBaseDataset
from .datasets import TimeSeriesDataset, RandomDataset, SineWaveDataset
from .data_providers import (
    DataProvider,
    RandomDataProvider,
    SineWaveDataProvider,
    InfluxDataProvider,
)
from .sensor_tag import SensorTag, normalize_sensor_tag, extract_tag_name
from .import_utils import import_location, BackCompatibleLocations
from .exceptions import (
    ConfigException,
    InsufficientDataError,
    SensorTagNormalizationError,
    NoSuitableDataProviderError,
)

__all__ = [
    "GordoBaseDataset",
    "TimeSeriesDataset",
    "RandomDataset",
    "SineWaveDataset",
    "DataProvider",
    "RandomDataProvider",
    "SineWaveDataProvider",
    "InfluxDataProvider",
    "SensorTag",
    "normalize_sensor_tag",
    "extract_tag_name",
    "import_location",
    "BackCompatibleLocations",
    "ConfigException",
    "InsufficientDataError",
    "SensorTagNormalizationError",
    "NoSuitableDataProviderError",
]
