from .metrics import GordoServerPrometheusMetrics, create_registry

__all__ = ["GordoServerPrometheusMetrics", "create_registry"]
