from .base import GordoBase
from .register import register_model_builder
from . import factories  # noqa: F401  (registers the factory kinds)
from .models import (
    TorchAutoEncoder,
    TorchBaseEstimator,
    TorchLSTMAutoEncoder,
    TorchLSTMForecast,
)

__all__ = [
    "GordoBase",
    "register_model_builder",
    "TorchBaseEstimator",
    "TorchAutoEncoder",
    "TorchLSTMAutoEncoder",
    "TorchLSTMForecast",
]
