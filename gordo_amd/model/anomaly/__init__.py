from .base import AnomalyDetectorBase
from .diff import DiffBasedAnomalyDetector

__all__ = ["AnomalyDetectorBase", "DiffBasedAnomalyDetector"]
