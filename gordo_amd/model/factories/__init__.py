from .feedforward_autoencoder import (
    feedforward_hourglass,
    feedforward_model,
    feedforward_symmetric,
)
from .lstm_autoencoder import lstm_hourglass, lstm_model, lstm_symmetric

__all__ = [
    "feedforward_model",
    "feedforward_symmetric",
    "feedforward_hourglass",
    "lstm_model",
    "lstm_symmetric",
    "lstm_hourglass",
]
