"""Misc utilities (spec: gordo/util/utils.py — capture_args)."""
from __future__ import annotations

import functools
import inspect


def capture_args(init):
    """
    Decorator for ``__init__`` storing the call's arguments into
    ``self._params`` (consumed by reporters / to_dict round-trips).

    >>> class A:
    ...     @capture_args
    ...     def __init__(self, x, y=2):
    ...         pass
    >>> A(1)._params
    {'x': 1, 'y': 2}
    """

    @functools.wraps(init)
    def wrapper(self, *args, **kwargs):
        sig = inspect.signature(init)
        bound = sig.bind(self, *args, **kwargs)
        bound.apply_defaults()
        params = dict(bound.arguments)
        params.pop("self", None)
        # flatten the VAR_KEYWORD parameter whatever its name
        for pname, p in sig.parameters.items():
            if p.kind is inspect.Parameter.VAR_KEYWORD and pname in params:
                extra = params.pop(pname)
                if isinstance(extra, dict):
                    params.update(extra)
        self._params = params
        return init(self, *args, **kwargs)

    return wrapper
