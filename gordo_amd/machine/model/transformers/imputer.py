"""
InfImputer — replace ±inf (spec: gordo/machine/model/transformers/imputer.py:12-127).
"""
from __future__ import annotations

from typing import Optional

import numpy as np
from sklearn.base import BaseEstimator, TransformerMixin


class InfImputer(BaseEstimator, TransformerMixin):
    """
    Fill ±inf values in a 2d array.

    strategy="minmax": per-feature observed min/max ∓/± a delta;
    strategy="extremes": the dtype's extreme values;
    or explicit ``inf_fill_value`` / ``neg_inf_fill_value``.
    """

    def __init__(
        self,
        inf_fill_value: Optional[float] = None,
        neg_inf_fill_value: Optional[float] = None,
        strategy: str = "minmax",
        delta: float = 2.0,
    ):
        self.inf_fill_value = inf_fill_value
        self.neg_inf_fill_value = neg_inf_fill_value
        self.strategy = strategy
        self.delta = delta

    def get_params(self, deep=True):
        return {
            "inf_fill_value": self.inf_fill_value,
            "neg_inf_fill_value": self.neg_inf_fill_value,
            "strategy": self.strategy,
            "delta": self.delta,
        }

    def fit(self, X, y=None):
        X = np.asarray(getattr(X, "values", X))
        if self.strategy == "minmax":
            masked = np.ma.masked_invalid(X.astype(np.float64))
            self._min_fill_values = masked.min(axis=0).filled(0.0) - self.delta
            self._max_fill_values = masked.max(axis=0).filled(0.0) + self.delta
        elif self.strategy == "extremes":
            finfo = np.finfo(X.dtype if X.dtype.kind == "f" else np.float64)
            self._min_fill_values = np.full(X.shape[1], finfo.min)
            self._max_fill_values = np.full(X.shape[1], finfo.max)
        else:
            raise ValueError(f"Unknown strategy {self.strategy!r}")
        return self

    def transform(self, X, y=None):
        X = np.asarray(getattr(X, "values", X)).copy()
        if self.inf_fill_value is not None:
            X[np.isposinf(X)] = self.inf_fill_value
        if self.neg_inf_fill_value is not None:
            X[np.isneginf(X)] = self.neg_inf_fill_value
        for col in range(X.shape[1]):
            values = X[:, col]
            values[np.isposinf(values)] = self._max_fill_values[col]
            values[np.isneginf(values)] = self._min_fill_values[col]
        return X
