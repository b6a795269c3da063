"""
Workflow-generation helpers: tz-enforcing YAML loading (unwrapping the
``spec.config`` CRD form), the jinja2 template environment with the
``yaml`` filter, and docker image-pull-policy selection.

Spec: gordo/workflow/workflow_generator/workflow_generator.py:60-134.
"""
from __future__ import annotations

import io
import os
from typing import Any, Dict, Union

import dateutil.parser
import jinja2
import yaml

from ...util.version import GordoRelease, GordoSHA, parse_version


def _timestamp_constructor(loader, node):
    parsed = dateutil.parser.parse(node.value)
    if parsed.tzinfo is None:
        raise ValueError(
            f"Timestamp {node.value!r} in config lacks timezone information"
        )
    return parsed


class _TzRequiredLoader(yaml.SafeLoader):
    pass


_TzRequiredLoader.add_constructor(
    "tag:yaml.org,2002:timestamp", _timestamp_constructor
)


def get_dict_from_yaml(config: Union[str, os.PathLike, io.IOBase]) -> Dict[str, Any]:
    """
    Load a config YAML (path, file object, or raw string) requiring
    timezones on timestamps; unwraps the ``spec.config`` CRD form.
    """
    if isinstance(config, io.IOBase):
        content = config.read()
    elif isinstance(config, os.PathLike) or (
        isinstance(config, str) and "\n" not in config and os.path.isfile(config)
    ):
        with open(config) as f:
            content = f.read()
    else:
        content = config
    data = yaml.load(content, Loader=_TzRequiredLoader)
    if not isinstance(data, dict):
        raise ValueError("Config YAML must be a mapping")
    # CRD form: {apiVersion..., spec: {config: {...}}}
    if "machines" not in data and "spec" in data:
        spec = data.get("spec") or {}
        if isinstance(spec, dict) and "config" in spec:
            data = spec["config"]
    return data


def _yaml_filter(value, indent: int = 0) -> str:
    dumped = yaml.safe_dump(value, default_flow_style=False)
    pad = " " * indent
    return "\n".join(
        pad + line if i else line
        for i, line in enumerate(dumped.rstrip("\n").split("\n"))
    )


def load_workflow_template(workflow_template: Union[str, os.PathLike]) -> jinja2.Template:
    """Load a jinja2 workflow template with the ``yaml`` filter
    available."""
    path = os.fspath(workflow_template)
    env = jinja2.Environment(
        loader=jinja2.FileSystemLoader(os.path.dirname(path) or "."),
        undefined=jinja2.StrictUndefined,
        trim_blocks=True,
        lstrip_blocks=True,
    )
    env.filters["yaml"] = _yaml_filter
    return env.get_template(os.path.basename(path))


def default_image_pull_policy(gordo_version: str) -> str:
    """``Always`` for mutable tags (latest/PR/major-only releases),
    ``IfNotPresent`` for exact releases and SHAs
    (spec: workflow_generator.py:126-134)."""
    version = parse_version(gordo_version)
    if isinstance(version, GordoSHA):
        return "IfNotPresent"
    if isinstance(version, GordoRelease):
        if version.only_major() or version.only_major_minor():
            return "Always"
        return "IfNotPresent"
    return "Always"
