from .base import base_blueprint
from .anomaly import anomaly_blueprint

__all__ = ["base_blueprint", "anomaly_blueprint"]
