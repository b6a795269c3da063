"""
from_definition — recursive YAML-dict → live object graph.

Behavioral spec (gordo/serializer/from_definition.py):
  * A dict with a single key that looks like an import path is an
    object: ``{"sklearn.pipeline.Pipeline": {...kwargs...}}``. The
    kwargs dict (or None) is recursively resolved and passed to the
    constructor.
  * ``Pipeline``/``FeatureUnion`` ``steps``/``transformer_list`` get
    special handling: each element may itself be a definition, a bare
    import-path string, or a ``[name, definition]`` pair.
  * A bare string that imports as a class is instantiated with no
    args in steps context; elsewhere strings that import resolve to
    the imported object (class or function) when the import succeeds.
  * A class with a ``from_definition`` classmethod hook is built via
    that hook (reference :190-191).
  * Tuple-annotated constructor params receive list→tuple coercion
    (reference create_instance:78-110).
"""
from __future__ import annotations

import inspect
import logging
from typing import Any, Dict, List, Union

from ..core.import_utils import import_location
from .utils import is_tuple_type

logger = logging.getLogger(__name__)

__all__ = ["from_definition", "load_params_from_definition", "build_callbacks"]


def _looks_like_import_path(s: str) -> bool:
    if not isinstance(s, str) or "." not in s:
        return False
    head = s.split(".")[0]
    return head.isidentifier() and not s.endswith(".")


def _maybe_import(s: str):
    """Import a dotted-path string, or return the string unchanged."""
    if not _looks_like_import_path(s):
        return s
    try:
        return import_location(s)
    except (ImportError, AttributeError, ValueError):
        return s


def _resolve_steps(steps: List[Any]) -> List[Any]:
    """Build a sklearn Pipeline ``steps`` list from definitions."""
    out = []
    for i, step in enumerate(steps):
        if isinstance(step, (list, tuple)) and len(step) == 2 and isinstance(
            step[0], str
        ) and not _looks_like_import_path(step[0]):
            name, obj = step[0], _resolve_value(step[1], instantiate_strings=True)
            out.append((name, obj))
            continue
        obj = _resolve_value(step, instantiate_strings=True)
        name = f"step_{i}"
        cls_name = type(obj).__name__ if not inspect.isclass(obj) else obj.__name__
        out.append((f"{name}_{cls_name}", obj))
    return out


_STEP_LIST_KEYS = {"steps", "transformer_list"}


def create_instance(cls, **params):
    """Instantiate ``cls`` coercing list-valued args whose annotation is
    a tuple type (YAML has no tuples)."""
    try:
        sig = inspect.signature(cls.__init__)
    except (TypeError, ValueError):
        sig = None
    if sig is not None:
        for name, value in list(params.items()):
            if isinstance(value, list) and name in sig.parameters:
                param = sig.parameters[name]
                ann = param.annotation
                # coerce when the annotation OR the default says tuple
                # (YAML/JSON have no tuples; sklearn validates strictly)
                if (
                    ann is not inspect.Parameter.empty and is_tuple_type(ann)
                ) or isinstance(param.default, tuple):
                    params[name] = tuple(value)
    return cls(**params)


def _resolve_params(cls, params: Dict[str, Any]) -> Dict[str, Any]:
    resolved: Dict[str, Any] = {}
    for key, value in params.items():
        if key in _STEP_LIST_KEYS and isinstance(value, list):
            resolved[key] = _resolve_steps(value)
        else:
            resolved[key] = _resolve_value(value)
    return resolved


def _resolve_value(value: Any, instantiate_strings: bool = False) -> Any:
    if isinstance(value, dict):
        if len(value) == 1:
            key = next(iter(value))
            if _looks_like_import_path(key):
                obj = _maybe_import(key)
                if not isinstance(obj, str):
                    return _construct(obj, value[key])
        # plain dict: resolve values
        return {k: _resolve_value(v) for k, v in value.items()}
    if isinstance(value, list):
        return [_resolve_value(v) for v in value]
    if isinstance(value, str) and _looks_like_import_path(value):
        obj = _maybe_import(value)
        if isinstance(obj, str):
            return obj
        if inspect.isclass(obj):
            # Estimator classes given as bare strings are instantiated with
            # defaults (reference gordo/serializer/from_definition.py:290-303:
            # `{"base_estimator": "sklearn...RandomForestRegressor"}` →
            # RandomForestRegressor()); other classes stay classes unless the
            # caller asked for instantiation.
            from sklearn.base import BaseEstimator

            if instantiate_strings or issubclass(obj, BaseEstimator):
                return _construct(obj, None)
        return obj
    return value


def _construct(obj: Any, params: Union[None, Dict[str, Any], List[Any]]) -> Any:
    """Instantiate ``obj`` (class/callable) from its params definition."""
    if hasattr(obj, "from_definition") and inspect.ismethod(
        getattr(obj, "from_definition", None)
    ):
        return obj.from_definition(params if params is not None else {})
    if params is None:
        return obj() if inspect.isclass(obj) or callable(obj) else obj
    if isinstance(params, dict):
        resolved = _resolve_params(obj, params)
        return create_instance(obj, **resolved)
    if isinstance(params, list):
        return obj(_resolve_value(params))
    return obj(params)


def from_definition(definition: Dict[str, Any]):
    """
    Build a live object (usually a sklearn Pipeline) from a config
    definition dict.

    Examples
    --------
    >>> model = from_definition(
    ...     {"sklearn.pipeline.Pipeline": {
    ...         "steps": ["sklearn.preprocessing.MinMaxScaler",
    ...                   {"sklearn.decomposition.PCA": {"n_components": 2}}]}})
    >>> type(model).__name__
    'Pipeline'
    >>> model.steps[1][1].n_components
    2
    """
    if not isinstance(definition, dict) or len(definition) != 1:
        raise ValueError(
            "Model definition must be a single-key dict of "
            f"{{import.path: params}}; got {definition!r}"
        )
    key = next(iter(definition))
    obj = _maybe_import(key)
    if isinstance(obj, str):
        raise ImportError(f"Cannot import model class {key!r}")
    return _construct(obj, definition[key])


def load_params_from_definition(definition: Dict[str, Any]) -> Dict[str, Any]:
    """Resolve a params dict (values may contain nested definitions)
    without treating the top level as an object definition."""
    if not isinstance(definition, dict):
        raise ValueError("Expected a dict of params")
    return _resolve_params(None, definition)


def build_callbacks(definitions: List[Any]) -> List[Any]:
    """Build a list of callback objects from definitions
    (reference from_definition.py:337-373)."""
    return [_resolve_value(d, instantiate_strings=True) for d in definitions]
