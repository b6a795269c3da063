"""
Model-factory registry (spec: gordo/machine/model/register.py:10-75):
``{estimator_type: {kind_name: builder_fn}}`` maintained as a
class-level dict on the decorator; builders must accept ``n_features``.
"""
from __future__ import annotations

import inspect
from typing import Callable, Dict


class register_model_builder:
    """
    Decorator registering a model-builder function under an estimator
    type.

    >>> @register_model_builder(type="KerasAutoEncoder")
    ... def my_special_model(n_features, **kwargs):
    ...     pass
    >>> "my_special_model" in register_model_builder.factories["KerasAutoEncoder"]
    True
    """

    factories: Dict[str, Dict[str, Callable]] = {}

    def __init__(self, type: str):
        self.type = type

    def __call__(self, build_fn: Callable) -> Callable:
        self._validate_func(build_fn)
        self.factories.setdefault(self.type, {})[build_fn.__name__] = build_fn
        return build_fn

    @staticmethod
    def _validate_func(func: Callable):
        params = inspect.signature(func).parameters
        if "n_features" not in params:
            raise ValueError(
                f"Model builder {func.__name__} must accept 'n_features'"
            )
