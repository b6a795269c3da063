"""AnomalyDetectorBase (spec: gordo/machine/model/anomaly/base.py:11-23)."""
from __future__ import annotations

import abc
from datetime import timedelta
from typing import Optional, Union

import numpy as np
import pandas as pd

from sklearn.base import BaseEstimator

from ..base import GordoBase


class AnomalyDetectorBase(BaseEstimator, GordoBase, metaclass=abc.ABCMeta):
    @abc.abstractmethod
    def anomaly(
        self,
        X: Union[pd.DataFrame, np.ndarray],
        y: Union[pd.DataFrame, np.ndarray],
        frequency: Optional[timedelta] = None,
    ) -> pd.DataFrame:
        """Score the anomaly between X's predictions and y."""
        ...
