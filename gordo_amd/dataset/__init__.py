from .base import GordoBaseDataset, InsufficientDataError, dataset_from_config
from .datasets import RandomDataset, TimeSeriesDataset

__all__ = [
    "GordoBaseDataset",
    "InsufficientDataError",
    "dataset_from_config",
    "TimeSeriesDataset",
    "RandomDataset",
]
