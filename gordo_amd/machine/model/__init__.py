from .base import GordoBase
from .models import (
    KerasBaseEstimator,
    KerasAutoEncoder,
    KerasRawModelRegressor,
    KerasLSTMBaseEstimator,
    KerasLSTMAutoEncoder,
    KerasLSTMForecast,
    create_keras_timeseriesgenerator,
)
from .register import register_model_builder
from . import factories  # noqa: F401  (registers the builders)

__all__ = [
    "GordoBase",
    "KerasBaseEstimator",
    "KerasAutoEncoder",
    "KerasRawModelRegressor",
    "KerasLSTMBaseEstimator",
    "KerasLSTMAutoEncoder",
    "KerasLSTMForecast",
    "create_keras_timeseriesgenerator",
    "register_model_builder",
]
