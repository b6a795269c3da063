"""Model output helper (spec: gordo/server/model_io.py:16-40)."""
from __future__ import annotations

import logging

import numpy as np

logger = logging.getLogger(__name__)


def get_model_output(model, X: np.ndarray) -> np.ndarray:
    """Predict, falling back to transform. With GORDO_SERVE_BATCH=1,
    concurrent predictions for the same model coalesce through the
    micro-batcher (server/batcher.py)."""
    try:
        from . import batcher

        if batcher.enabled() and hasattr(model, "predict"):
            return batcher.batched_predict(model, X)
        return model.predict(X)
    except AttributeError:
        try:
            return model.transform(X)
        except Exception as exc:
            logger.error("Failed to predict or transform; error: %s", exc)
            raise
