from .base import AnomalyDetectorBase
from .diff import DiffBasedAnomalyDetector, DiffBasedKFCVAnomalyDetector

__all__ = [
    "AnomalyDetectorBase",
    "DiffBasedAnomalyDetector",
    "DiffBasedKFCVAnomalyDetector",
]
