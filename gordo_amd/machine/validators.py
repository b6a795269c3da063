"""
Descriptor-based Machine attribute validation
(spec: gordo/machine/validators.py).
"""
from __future__ import annotations

import datetime
import logging
import re

import pandas as pd

logger = logging.getLogger(__name__)


class BaseDescriptor:
    """Attribute descriptor validating on assignment."""

    def __set_name__(self, owner, name):
        self.name = name

    def __get__(self, instance, owner):
        if instance is None:
            return self
        return instance.__dict__.get(self.name)

    def __set__(self, instance, value):
        self.validate(value)
        instance.__dict__[self.name] = value

    def validate(self, value):
        raise NotImplementedError()


class ValidDataset(BaseDescriptor):
    def validate(self, value):
        from ..core.base import GordoBaseDataset

        if not isinstance(value, GordoBaseDataset):
            raise TypeError(
                f"dataset must be a GordoBaseDataset; got {type(value)}"
            )


class ValidDatasetKwargs(BaseDescriptor):
    def validate(self, value):
        if not isinstance(value, dict):
            raise ValueError(f"Expected a dict, got {type(value)}")
        self._validate_resolution(value)

    @staticmethod
    def _validate_resolution(value: dict):
        if "resolution" in value:
            try:
                pd.tseries.frequencies.to_offset(value["resolution"])
            except ValueError as e:
                raise ValueError(
                    f"Invalid pandas frequency {value['resolution']!r}"
                ) from e


class ValidModel(BaseDescriptor):
    """The model definition must be buildable via
    serializer.from_definition (the reference runs a full dry-run build:
    gordo/machine/validators.py:86-92)."""

    def validate(self, value):
        if not isinstance(value, dict):
            raise ValueError(f"model must be a dict definition; got {type(value)}")
        from ..serializer import from_definition

        try:
            from_definition(value)
        except Exception as e:
            raise ValueError(f"Invalid model definition: {e}") from e


class ValidMetadata(BaseDescriptor):
    def validate(self, value):
        from .metadata import Metadata

        if value is not None and not isinstance(value, (dict, Metadata)):
            raise ValueError(f"metadata must be dict or Metadata; got {type(value)}")


class ValidDatetime(BaseDescriptor):
    def validate(self, value):
        if not isinstance(value, (datetime.datetime, pd.Timestamp)):
            raise ValueError(f"Expected datetime, got {type(value)}")
        if value.tzinfo is None:
            raise ValueError(f"Datetime {value} must be timezone-aware")


class ValidTagList(BaseDescriptor):
    def validate(self, value):
        if not isinstance(value, (list, tuple)) or len(value) == 0:
            raise ValueError("Requires a non-empty list of tags")


class ValidUrlString(BaseDescriptor):
    """k8s/DNS-compatible name: lowercase alphanumerics and dashes,
    max 63 chars (reference validators.py:269)."""

    def validate(self, value):
        if not isinstance(value, str):
            raise ValueError(f"Expected a string, got {type(value)}")
        if not self.valid_url_string(value):
            raise ValueError(
                f"{value!r} is not a valid name: lowercase alphanumerics "
                "and dashes only, must not start/end with a dash"
            )
        if len(value) > 63:
            raise ValueError(f"Name {value!r} exceeds 63 characters")

    @staticmethod
    def valid_url_string(value: str) -> bool:
        """
        >>> ValidUrlString.valid_url_string("my-model-01")
        True
        >>> ValidUrlString.valid_url_string("My_Model")
        False
        """
        return bool(re.fullmatch(r"[a-z0-9]([a-z0-9\-]*[a-z0-9])?", value))


def fix_resource_limits(resources: dict) -> dict:
    """Ensure limits >= requests for cpu/memory in a k8s resources dict
    (reference validators.py:173)."""
    requests = resources.get("requests", {}) or {}
    limits = resources.get("limits", {}) or {}
    for key in ("cpu", "memory"):
        req, lim = requests.get(key), limits.get(key)
        if req is not None and lim is not None:
            # values must be plain ints (the unit scale is fixed by the
            # config schema); reference raises rather than guessing at
            # k8s quantity strings (validators.py:173 there)
            try:
                req_i, lim_i = int(req), int(lim)
            except (TypeError, ValueError) as e:
                raise ValueError(
                    f"resource {key!r} values must be ints; got "
                    f"requests={req!r} limits={lim!r}"
                ) from e
            if req_i > lim_i:
                limits[key] = req_i
    if limits:
        resources["limits"] = limits
    return resources


def fix_runtime(runtime: dict) -> dict:
    """Apply fix_resource_limits to each known runtime pod section."""
    runtime = dict(runtime or {})
    for section_name in ("server", "builder", "client", "influx", "prometheus_metrics_server"):
        section = runtime.get(section_name)
        if isinstance(section, dict) and "resources" in section:
            section["resources"] = fix_resource_limits(section["resources"])
    return runtime


class ValidMachineRuntime(BaseDescriptor):
    def validate(self, value):
        if not isinstance(value, dict):
            raise ValueError(f"runtime must be a dict; got {type(value)}")


class ValidStrList(BaseDescriptor):
    def validate(self, value):
        if not isinstance(value, list) or not all(isinstance(v, str) for v in value):
            raise ValueError("Expected a list of strings")
