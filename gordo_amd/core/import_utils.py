"""
Dotted-path import machinery used by the serializer and dataset layer.

Spec: gordo_core's ``import_location`` as consumed by the reference
serializer (gordo/serializer/from_definition.py:179-181).

Paths beginning with ``gordo.`` or ``gordo_core.`` are transparently
aliased into this package so that reference YAML configs are drop-in
compatible.
"""
from __future__ import annotations

import importlib
from typing import Any, Dict, Optional, Tuple

# Back-compatible location aliases: reference configs name classes under
# the ``gordo``/``gordo_core`` packages; we serve them from gordo_amd.
BackCompatibleLocations: Dict[str, str] = {
    "gordo.": "gordo_amd.",
    "gordo_core.base.": "gordo_amd.core.base.",
    "gordo_core.time_series.": "gordo_amd.core.datasets.",
    "gordo_core.data_providers.": "gordo_amd.core.data_providers.",
    "gordo_core.": "gordo_amd.core.",
    # tensorflow/keras callbacks etc. have no analog here; no alias.
}


def resolve_alias(import_path: str) -> str:
    for prefix, replacement in BackCompatibleLocations.items():
        if import_path.startswith(prefix):
            return replacement + import_path[len(prefix):]
    return import_path


def split_location(import_path: str) -> Tuple[Optional[str], str]:
    if "." in import_path:
        module_path, attr = import_path.rsplit(".", 1)
        return module_path, attr
    return None, import_path


def import_location(import_path: str) -> Any:
    """
    Import an object by dotted path, e.g.
    ``"sklearn.preprocessing.MinMaxScaler"``.

    >>> import_location("sklearn.preprocessing.MinMaxScaler").__name__
    'MinMaxScaler'
    """
    import_path = resolve_alias(import_path)
    module_path, attr = split_location(import_path)
    if module_path is None:
        raise ImportError(f"{import_path!r} is not a dotted import path")
    try:
        module = importlib.import_module(module_path)
    except ImportError:
        # maybe the attr is a nested class: a.b.C.D
        parent_path, parent_attr = split_location(module_path)
        if parent_path is None:
            raise
        parent = import_location(module_path)
        return getattr(parent, attr)
    try:
        return getattr(module, attr)
    except AttributeError as e:
        raise ImportError(
            f"Module {module_path!r} has no attribute {attr!r}"
        ) from e
