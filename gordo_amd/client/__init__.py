from .client import Client
from .forwarders import ForwardPredictionsIntoInflux, ForwardPredictionsToDisk
from .utils import EndpointMetadata, PredictionResult, make_date_ranges

__all__ = [
    "Client",
    "PredictionResult",
    "EndpointMetadata",
    "make_date_ranges",
    "ForwardPredictionsIntoInflux",
    "ForwardPredictionsToDisk",
]
