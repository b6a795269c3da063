"""
GordoBaseDataset — the dataset abstraction.

Spec (reference import sites): ``GordoBaseDataset.from_dict / to_dict /
get_data / get_metadata`` used at gordo/builder/build_model.py:208-215
and gordo/machine/machine.py:190-195.
"""
from __future__ import annotations

from typing import Any, Dict, Tuple

import pandas as pd

from .import_utils import import_location


class GordoBaseDataset:
    _params: Dict[str, Any]

    @classmethod
    def from_dict(cls, config: Dict[str, Any]) -> "GordoBaseDataset":
        from . import datasets  # local import to avoid cycle

        config = dict(config or {})
        kind = config.pop("type", "TimeSeriesDataset")
        if "." in kind:
            dataset_cls = import_location(kind)
        else:
            dataset_cls = getattr(datasets, kind, None)
            if dataset_cls is None:
                from .exceptions import ConfigException

                raise ConfigException(f"No dataset type named {kind!r}")
        return dataset_cls(**config)

    def to_dict(self) -> Dict[str, Any]:
        d = dict(self._params)
        d["type"] = type(self).__name__
        return d

    def get_data(self) -> Tuple[pd.DataFrame, pd.DataFrame]:
        """Return (X, y) as time-indexed DataFrames."""
        raise NotImplementedError()

    def get_metadata(self) -> Dict[str, Any]:
        return {}

    def __repr__(self):
        return f"{type(self).__name__}({self._params!r})"

    def __eq__(self, other):
        return type(self) is type(other) and self.to_dict() == other.to_dict()
