from .feedforward_autoencoder import (
    feedforward_model,
    feedforward_symmetric,
    feedforward_hourglass,
)
from .lstm_autoencoder import lstm_model, lstm_symmetric, lstm_hourglass

__all__ = [
    "feedforward_model",
    "feedforward_symmetric",
    "feedforward_hourglass",
    "lstm_model",
    "lstm_symmetric",
    "lstm_hourglass",
]
