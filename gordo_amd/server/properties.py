"""
Request-scoped model properties recovered from stored metadata
(spec: gordo/server/properties.py).
"""
from __future__ import annotations

import copy
from typing import Any, Dict, List, Optional

import pandas as pd
from flask import g

from ..core.sensor_tag import SensorTag
from ..utils import normalize_sensor_tags


def find_path_in_dict(path: List[str], data: dict) -> Any:
    """
    >>> find_path_in_dict(["parent", "child"], {"parent": {"child": 42}})
    42
    """
    remaining = copy.copy(path)
    remaining.reverse()
    curr = data
    while remaining:
        key = remaining.pop()
        if key not in curr:
            exception_path = ".".join(path[: len(path) - len(remaining)])
            raise KeyError("'%s' is absent" % exception_path)
        curr = curr[key]
    return curr


def get_frequency():
    """The resolution the model's dataset was trained with."""
    return pd.tseries.frequencies.to_offset(g.metadata["dataset"]["resolution"])


def load_build_dataset_metadata() -> dict:
    try:
        return find_path_in_dict(
            ["metadata", "build_metadata", "dataset"], g.metadata
        )
    except KeyError as e:
        raise ValueError("Unable to load build dataset metadata: %s" % str(e))


def get_normalize_additional_fields(dataset: Dict[str, Any]) -> Dict[str, Optional[str]]:
    if dataset.get("default_tag"):
        return dataset["default_tag"]
    if dataset.get("asset"):
        return {"asset": dataset["asset"]}
    return {}


def get_tags() -> List[SensorTag]:
    dataset = g.metadata["dataset"]
    tag_list = dataset["tag_list"]
    build_dataset_metadata = load_build_dataset_metadata()
    additional_fields = get_normalize_additional_fields(dataset)
    return normalize_sensor_tags(build_dataset_metadata, tag_list, **additional_fields)


def get_target_tags() -> List[SensorTag]:
    orig = g.metadata["dataset"].get("target_tag_list") or []
    if orig:
        build_dataset_metadata = load_build_dataset_metadata()
        additional_fields = get_normalize_additional_fields(g.metadata["dataset"])
        return normalize_sensor_tags(build_dataset_metadata, orig, **additional_fields)
    return get_tags()
