"""
MlFlowReporter — push build metadata as MLflow metrics/params
(behavioral spec: gordo/reporters/mlflow.py — metadata flattened into
≤200-metric / ≤100-param batches, run keyed by the builder cache key).

The AzureML-workspace auth path of the reference requires azureml-sdk
and network access, neither present here; the reporter supports any
reachable MLflow tracking URI and degrades with a clear error when
mlflow is not importable.
"""
from __future__ import annotations

import logging
from contextlib import contextmanager
from typing import Any, Dict, List, Tuple

from .base import BaseReporter
from .exceptions import ReporterException
from ..util.utils import capture_args

logger = logging.getLogger(__name__)

try:
    import mlflow  # noqa: F401

    HAS_MLFLOW = True
except ImportError:
    HAS_MLFLOW = False


class MlflowLoggingError(ReporterException):
    pass


def _flatten(d: Dict[str, Any], prefix: str = "") -> List[Tuple[str, Any]]:
    out: List[Tuple[str, Any]] = []
    for k, v in (d or {}).items():
        key = f"{prefix}.{k}" if prefix else str(k)
        if isinstance(v, dict):
            out.extend(_flatten(v, key))
        else:
            out.append((key, v))
    return out


def get_machine_log_items(machine) -> Tuple[List[Tuple[str, float]], List[Tuple[str, str]]]:
    """Split a machine's build metadata into (metrics, params):
    numeric leaves become metrics, the rest params."""
    metrics: List[Tuple[str, float]] = []
    params: List[Tuple[str, str]] = []
    flat = _flatten(machine.to_dict().get("metadata", {}))
    for key, value in flat:
        if isinstance(value, bool):
            params.append((key, str(value)))
        elif isinstance(value, (int, float)):
            metrics.append((key, float(value)))
        else:
            params.append((key, str(value)[:250]))
    return metrics, params


def batch_log_items(items: List, batch_size: int) -> List[List]:
    """
    >>> batch_log_items(list(range(5)), 2)
    [[0, 1], [2, 3], [4]]
    """
    return [items[i : i + batch_size] for i in range(0, len(items), batch_size)]


@contextmanager
def mlflow_context(run_name: str, model_key: str, tracking_uri: str = None):
    if not HAS_MLFLOW:
        raise MlflowLoggingError("mlflow is not installed in this environment")
    if tracking_uri:
        mlflow.set_tracking_uri(tracking_uri)
    run = mlflow.start_run(run_name=f"{run_name}-{model_key[:8]}")
    try:
        yield mlflow
    finally:
        mlflow.end_run()


class MlFlowReporter(BaseReporter):
    MAX_METRICS_PER_BATCH = 200
    MAX_PARAMS_PER_BATCH = 100

    @capture_args
    def __init__(self, tracking_uri: str = None, **kwargs):
        self.tracking_uri = tracking_uri
        self.kwargs = kwargs

    def report(self, machine):
        if not HAS_MLFLOW:
            raise MlflowLoggingError("mlflow is not installed in this environment")
        from ..builder import ModelBuilder

        cache_key = ModelBuilder(machine).cache_key
        metrics, params = get_machine_log_items(machine)
        with mlflow_context(machine.name, cache_key, self.tracking_uri) as mf:
            for batch in batch_log_items(metrics, self.MAX_METRICS_PER_BATCH):
                mf.log_metrics({k: v for k, v in batch})
            for batch in batch_log_items(params, self.MAX_PARAMS_PER_BATCH):
                mf.log_params({k: v for k, v in batch})
