"""
into_definition — live object graph → config definition dict (the
inverse of ``from_definition``).

Behavioral spec (gordo/serializer/into_definition.py): decompose via
``get_params(deep=False)``, recursing through nested estimators,
lists of (name, estimator) step pairs, classes and callables; optional
pruning of params that equal the constructor defaults.
"""
from __future__ import annotations

import inspect
import logging
from typing import Any, Dict

logger = logging.getLogger(__name__)

__all__ = ["into_definition"]


def _object_path(obj) -> str:
    cls = obj if inspect.isclass(obj) or inspect.isfunction(obj) else type(obj)
    return f"{cls.__module__}.{cls.__qualname__}"


def _default_params(obj) -> Dict[str, Any]:
    try:
        sig = inspect.signature(type(obj).__init__)
    except (TypeError, ValueError):
        return {}
    return {
        k: p.default
        for k, p in sig.parameters.items()
        if p.default is not inspect.Parameter.empty
    }


def _decompose_node(value: Any, prune_default_params: bool) -> Any:
    # primitives
    if value is None or isinstance(value, (bool, int, float, str)):
        return value
    # classes / functions referenced by value
    if inspect.isclass(value) or inspect.isfunction(value):
        return _object_path(value)
    # numpy scalars
    if hasattr(value, "item") and not hasattr(value, "get_params") and getattr(
        value, "shape", None
    ) == ():
        return value.item()
    if isinstance(value, (list, tuple)):
        decomposed = []
        for elem in value:
            # (name, estimator) step pairs → keep just the estimator definition
            if (
                isinstance(elem, tuple)
                and len(elem) == 2
                and isinstance(elem[0], str)
                and hasattr(elem[1], "get_params")
            ):
                decomposed.append(_decompose_node(elem[1], prune_default_params))
            else:
                decomposed.append(_decompose_node(elem, prune_default_params))
        return decomposed
    if isinstance(value, dict):
        return {k: _decompose_node(v, prune_default_params) for k, v in value.items()}
    # estimators / anything with sklearn get_params
    if hasattr(value, "get_params"):
        params = value.get_params(deep=False)
        if prune_default_params:
            defaults = _default_params(value)
            params = {
                k: v
                for k, v in params.items()
                if not (k in defaults and defaults[k] == v)
            }
        # drop memory/verbose-style None noise only when pruning is off? keep all
        return {
            _object_path(value): {
                k: _decompose_node(v, prune_default_params) for k, v in params.items()
            }
        }
    if callable(value):
        return _object_path(value)
    # fallback: repr-able objects with a to_dict
    if hasattr(value, "to_dict"):
        return value.to_dict()
    logger.warning("into_definition: cannot decompose %r; using repr", type(value))
    return repr(value)


def into_definition(pipeline, prune_default_params: bool = False) -> Dict[str, Any]:
    """
    Convert an estimator (e.g. a Pipeline) into its config-definition
    dict such that ``from_definition(into_definition(p))`` rebuilds an
    equivalent object.

    Examples
    --------
    >>> from sklearn.pipeline import Pipeline
    >>> from sklearn.preprocessing import MinMaxScaler
    >>> d = into_definition(Pipeline([("mms", MinMaxScaler())]),
    ...                     prune_default_params=True)
    >>> list(d)
    ['sklearn.pipeline.Pipeline']
    """
    return _decompose_node(pipeline, prune_default_params)
