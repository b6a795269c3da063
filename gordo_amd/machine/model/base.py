"""GordoBase — the model ABC (spec: gordo/machine/model/base.py:10)."""
from __future__ import annotations

import abc
from typing import Any, Dict, Optional

import numpy as np


class GordoBase(abc.ABC):
    @abc.abstractmethod
    def __init__(self, **kwargs):
        ...

    @abc.abstractmethod
    def get_metadata(self) -> Dict[str, Any]:
        """Return model metadata (e.g. training history)."""
        ...

    def score(
        self,
        X: np.ndarray,
        y: np.ndarray,
        sample_weight: Optional[np.ndarray] = None,
    ) -> float:
        raise NotImplementedError()
