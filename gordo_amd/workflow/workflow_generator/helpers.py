"""
patch_dict — merge a patch dict into a base dict with
add-or-change-only semantics (never removes keys).

Spec: gordo/workflow/workflow_generator/helpers.py:16-45.
"""
from __future__ import annotations

from copy import deepcopy
from typing import Any, Dict


def patch_dict(original: Dict[str, Any], patch: Dict[str, Any]) -> Dict[str, Any]:
    """
    Recursively overlay ``patch`` onto ``original``: keys present only
    in ``original`` are kept, keys in ``patch`` are added or replace,
    nested dicts merge recursively. Neither input is mutated.

    >>> patch_dict({"a": {"b": 1, "c": 2}}, {"a": {"c": 3, "d": 4}, "e": 5}) == \
        {"a": {"b": 1, "c": 3, "d": 4}, "e": 5}
    True
    """
    result = deepcopy(dict(original or {}))
    for key, value in (patch or {}).items():
        if (
            key in result
            and isinstance(result[key], dict)
            and isinstance(value, dict)
        ):
            result[key] = patch_dict(result[key], value)
        else:
            result[key] = deepcopy(value)
    return result
