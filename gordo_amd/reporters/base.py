"""Reporter base (spec: gordo/reporters/base.py:9-33)."""
from __future__ import annotations

import abc


class BaseReporter(abc.ABC):
    @abc.abstractmethod
    def report(self, machine):
        ...

    def to_dict(self) -> dict:
        from ..serializer import into_definition

        return into_definition(self)

    @classmethod
    def from_dict(cls, config: dict) -> "BaseReporter":
        from ..serializer import from_definition

        return from_definition(config)

    def get_params(self, deep=False) -> dict:
        return dict(getattr(self, "_params", {}))
