from .client import (
    Client,
    HttpUnprocessableEntity,
    ResourceGone,
    NotFound,
    BadGordoRequest,
)
from .forwarders import (
    PredictionForwarder,
    ForwardPredictionsIntoInflux,
    ForwardPredictionsToDisk,
)

__all__ = [
    "Client",
    "HttpUnprocessableEntity",
    "ResourceGone",
    "NotFound",
    "BadGordoRequest",
    "PredictionForwarder",
    "ForwardPredictionsIntoInflux",
    "ForwardPredictionsToDisk",
]
