"""
Machine-config loaders: parse raw YAML dicts whose ``model``,
``dataset``, ``evaluation``, ``metadata`` and ``runtime`` fields may be
nested YAML strings (``|`` blocks).

Spec: gordo/machine/loader.py:55-116 + gordo/machine/constants.py.
"""
from __future__ import annotations

from copy import deepcopy
from typing import Any, Dict

import yaml

from .constants import MACHINE_YAML_FIELDS
from ..core.exceptions import ConfigException


class MachineConfigException(ConfigException):
    pass


def _parse_nested_yaml(config: Dict[str, Any], fields) -> Dict[str, Any]:
    out = deepcopy(dict(config or {}))
    for field_name in fields:
        value = out.get(field_name)
        if isinstance(value, str):
            try:
                parsed = yaml.safe_load(value)
            except yaml.YAMLError as e:
                raise MachineConfigException(
                    f"Malformed nested YAML in field {field_name!r}: {e}"
                ) from e
            out[field_name] = parsed
    return out


def load_globals_config(config: Dict[str, Any], base_path: str = "globals") -> Dict[str, Any]:
    """Parse the ``globals:`` section of a config."""
    if config is None:
        return {}
    if not isinstance(config, dict):
        raise MachineConfigException(f"{base_path} must be a mapping")
    return _parse_nested_yaml(config, MACHINE_YAML_FIELDS)


def load_machine_config(config: Dict[str, Any], base_path: str = "machine") -> Dict[str, Any]:
    """Parse a single machine's config section."""
    if not isinstance(config, dict):
        raise MachineConfigException(f"{base_path} must be a mapping")
    parsed = _parse_nested_yaml(config, MACHINE_YAML_FIELDS)
    if not parsed.get("name"):
        raise MachineConfigException(f"{base_path} requires a 'name'")
    return parsed


def load_model_config(config: Dict[str, Any], base_path: str = "machine") -> Dict[str, Any]:
    """Parse a machine config and require its ``model`` field."""
    parsed = load_machine_config(config, base_path)
    model = parsed.get("model")
    if not isinstance(model, dict):
        raise MachineConfigException(
            f"{base_path}.model must resolve to a mapping; got {type(model)}"
        )
    return parsed
