from .base import BaseReporter
from .exceptions import ReporterException
from .postgres import PostgresReporter
from .mlflow import MlFlowReporter

__all__ = ["BaseReporter", "ReporterException", "PostgresReporter", "MlFlowReporter"]
