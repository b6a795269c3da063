"""Resolve a custom ModelBuilder subclass by import path
(spec: gordo/builder/utils.py:8-17)."""
from __future__ import annotations

from typing import Optional, Type

from .build_model import ModelBuilder
from ..core.import_utils import import_location


def create_model_builder(model_builder_class: Optional[str]) -> Type[ModelBuilder]:
    if not model_builder_class:
        return ModelBuilder
    cls = import_location(model_builder_class)
    if not issubclass(cls, ModelBuilder):
        raise ValueError(f"{model_builder_class} is not a ModelBuilder subclass")
    return cls
