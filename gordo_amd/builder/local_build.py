"""
local_build — YAML string → trained (model, Machine) generator, the
in-process mini pipeline (spec: gordo/builder/local_build.py:14-70;
the test backbone).
"""
from __future__ import annotations

from typing import Iterable, Optional, Tuple

from .utils import create_model_builder
from ..workflow.workflow_generator.workflow_generator import get_dict_from_yaml
from ..workflow.config_elements.normalized_config import NormalizedConfig


def local_build(
    config_str: str,
    project_name: str = "local-build",
    model_builder_class: Optional[str] = None,
) -> Iterable[Tuple[object, "Machine"]]:
    """
    Build model(s) from a raw gordo config file string, in-process
    (no caching, no registry).

    Yields (model, machine) per machine in the config.
    """
    config = get_dict_from_yaml(config_str)
    norm = NormalizedConfig(config, project_name=project_name)
    builder_cls = create_model_builder(model_builder_class)
    for machine in norm.machines:
        yield builder_cls(machine=machine).build()
