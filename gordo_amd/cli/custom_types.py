"""Custom click parameter types (spec: gordo/cli/custom_types.py)."""
from __future__ import annotations

import ipaddress
import json
import re
from typing import Any, Tuple

import click
import yaml


class JSONParam(click.ParamType):
    """A JSON (or YAML) document parameter."""

    name = "json"

    def convert(self, value, param, ctx):
        if isinstance(value, dict):
            return value
        try:
            return json.loads(value)
        except (TypeError, ValueError):
            pass
        try:
            parsed = yaml.safe_load(value)
        except yaml.YAMLError:
            self.fail(f"{value!r} is not valid JSON/YAML", param, ctx)
            return
        if not isinstance(parsed, dict):
            self.fail(f"{value!r} did not parse to a mapping", param, ctx)
        return parsed


class REParam(click.ParamType):
    """A regular-expression parameter."""

    name = "regex"

    def convert(self, value, param, ctx):
        try:
            return re.compile(value)
        except re.error as e:
            self.fail(f"{value!r} is not a valid regex: {e}", param, ctx)


class HostIP(click.ParamType):
    """An IP address (or 0.0.0.0-style bind address)."""

    name = "host"

    def convert(self, value, param, ctx):
        try:
            ipaddress.ip_address(value)
            return value
        except ValueError:
            self.fail(f"{value!r} is not a valid IP address", param, ctx)


def key_value_par(val: str) -> Tuple[str, Any]:
    """Parse a 'key,value' pair used for --model-parameter.

    >>> key_value_par("a,2")
    ('a', 2)
    """
    key, _, value = val.partition(",")
    try:
        parsed = yaml.safe_load(value)
    except yaml.YAMLError:
        parsed = value
    return key, parsed


class IsoFormatDateTime(click.ParamType):
    """ISO-8601 datetime click param (tz-aware), e.g.
    2020-01-01T00:00:00+00:00."""

    name = "isodatetime"

    def convert(self, value, param, ctx):
        import dateutil.parser

        try:
            return dateutil.parser.isoparse(value)
        except ValueError:
            self.fail(f"{value!r} is not a valid ISO datetime", param, ctx)
