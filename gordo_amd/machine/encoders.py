"""JSON / YAML encoders for Machine round-trips
(spec: gordo/machine/encoders.py)."""
from __future__ import annotations

import datetime
import json

import numpy as np
import pandas as pd
import yaml

from ..core.sensor_tag import SensorTag

DATETIME_FORMAT = "%Y-%m-%dT%H:%M:%S%z"


class MachineJSONEncoder(json.JSONEncoder):
    def default(self, obj):
        if isinstance(obj, (datetime.datetime, pd.Timestamp)):
            return obj.isoformat()
        if isinstance(obj, SensorTag):
            return obj.to_json()
        if isinstance(obj, np.integer):
            return int(obj)
        if isinstance(obj, np.floating):
            return float(obj)
        if isinstance(obj, np.ndarray):
            return obj.tolist()
        return super().default(obj)


def multiline_str_representer(dumper, data):
    if "\n" in data:
        return dumper.represent_scalar("tag:yaml.org,2002:str", data, style="|")
    return dumper.represent_scalar("tag:yaml.org,2002:str", data)


class MachineSafeDumper(yaml.SafeDumper):
    pass


MachineSafeDumper.add_representer(str, multiline_str_representer)
MachineSafeDumper.add_representer(
    SensorTag,
    lambda dumper, tag: dumper.represent_dict(tag.to_json()),
)
MachineSafeDumper.add_representer(
    pd.Timestamp,
    lambda dumper, ts: dumper.represent_str(ts.isoformat()),
)
MachineSafeDumper.add_representer(
    datetime.datetime,
    lambda dumper, dt: dumper.represent_str(dt.isoformat()),
)
