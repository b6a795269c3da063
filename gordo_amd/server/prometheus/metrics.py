"""
Prometheus request metrics (behavioral spec:
gordo/server/prometheus/metrics.py — histogram
``gordo_server_request_duration_seconds`` + counter
``gordo_server_requests_total`` labeled (method, path, status_code,
model, version[, project]); the KEDA autoscaling query in the workflow
depends on these exact names).
"""
from __future__ import annotations

import timeit
from copy import copy
from http import HTTPStatus
from typing import Dict, Iterable, List, Optional, Tuple

from flask import Flask, Request, Response, g, request
from prometheus_client import Counter, Gauge, Histogram
from prometheus_client.registry import CollectorRegistry


def create_registry() -> CollectorRegistry:
    from prometheus_client.multiprocess import MultiProcessCollector

    registry = CollectorRegistry()
    try:
        MultiProcessCollector(registry)
    except Exception:
        # multiprocess mode needs PROMETHEUS_MULTIPROC_DIR; single-process
        # servers (werkzeug fallback) work with a plain registry.
        pass
    return registry


def to_status_code(response_status):
    if isinstance(response_status, HTTPStatus):
        return response_status.value
    return response_status


def url_rule_to_str(url_rule):
    return url_rule.rule if url_rule is not None else ""


def current_time():
    return timeit.default_timer()


class GordoServerPrometheusMetrics:
    """
    Request metrics for a Flask app.

    >>> from flask import Flask
    >>> from prometheus_client.registry import CollectorRegistry
    >>> app = Flask("test")
    >>> @app.route('/hello')
    ... def hello():
    ...     return 'Hello, World'
    >>> prometheus_metrics = GordoServerPrometheusMetrics(registry=CollectorRegistry())
    >>> prometheus_metrics.prepare_app(app)
    """

    prefix = "gordo_server"
    main_labels = ("method", "path", "status_code")

    @staticmethod
    def main_label_values(req: Request, resp: Response):
        return (
            req.method,
            url_rule_to_str(req.url_rule),
            to_status_code(resp.status_code),
        )

    def __init__(
        self,
        args_labels: Optional[Iterable[Tuple[str, str]]] = None,
        info: Optional[Dict[str, str]] = None,
        ignore_paths: Optional[Iterable[str]] = None,
        registry: Optional[CollectorRegistry] = None,
    ):
        self.args_labels = list(args_labels) if args_labels is not None else []
        self.ignore_paths = set(ignore_paths) if ignore_paths is not None else set()
        self.info = info
        self.label_names: List[str] = []
        self.label_values: List[str] = []
        self.args_names: List[str] = []

        if registry is None:
            registry = create_registry()
        self.registry = registry
        self.init_labels()
        self.request_duration_seconds = Histogram(
            f"{self.prefix}_request_duration_seconds",
            "HTTP request duration, in seconds",
            self.label_names,
            registry=registry,
        )
        self.request_count = Counter(
            f"{self.prefix}_requests_total",
            "Total HTTP requests",
            self.label_names,
            registry=registry,
        )

    def init_labels(self):
        label_names, label_values = [], []
        if self.info is not None:
            for name, value in self.info.items():
                label_names.append(name)
                label_values.append(value)
            gauge_info = Gauge(
                self.prefix + "_info",
                "Gordo information",
                label_names,
                registry=self.registry,
            )
            gauge_info.labels(*label_values).set(1)
        args_names = []
        for arg_name, label_name in self.args_labels:
            args_names.append(arg_name)
            label_names.append(label_name)
        self.args_names = args_names
        label_names.extend(self.main_labels)
        self.label_names = label_names
        self.label_values = label_values

    def request_label_values(self, req: Request, resp: Response):
        label_values = copy(self.label_values)
        view_args = req.view_args
        for arg_name in self.args_names:
            label_values.append(
                view_args.get(arg_name, "") if view_args is not None else ""
            )
        label_values.extend(self.main_label_values(req, resp))
        return label_values

    def prepare_gpu_gauges(self):
        """Per-GPU serving gauges (SURVEY §5.3/§5.5 'new framework'):
        models resident in the serving LRU and HBM bytes allocated —
        the capacity signals for a 288 GB MI355X serving thousands of
        models."""
        from prometheus_client import Gauge

        models_cached = Gauge(
            "gordo_server_models_cached",
            "Models resident in the serving LRU cache",
            registry=self.registry,
        )

        def _cache_size() -> float:
            from .. import utils as server_utils

            return float(server_utils.load_model.cache_info().currsize)

        models_cached.set_function(_cache_size)

        gpu_mem = Gauge(
            "gordo_server_gpu_memory_allocated_bytes",
            "torch HBM bytes allocated on the serving device",
            registry=self.registry,
        )

        def _gpu_mem() -> float:
            try:
                import torch

                if torch.cuda.is_available():
                    return float(torch.cuda.memory_allocated())
            except ImportError:
                pass
            return 0.0

        gpu_mem.set_function(_gpu_mem)

    def prepare_app(self, app: Flask):
        self.prepare_gpu_gauges()

        @app.before_request
        def _start_prometheus():
            g.prometheus_metrics = self
            g.prometheus_start_time = current_time()

        @app.after_request
        def _end_prometheus(response: Response) -> Response:
            url_rule = url_rule_to_str(request.url_rule)
            if url_rule in self.ignore_paths:
                return response
            label_values = self.request_label_values(request, response)
            self.request_duration_seconds.labels(*label_values).observe(
                current_time() - g.prometheus_start_time
            )
            self.request_count.labels(*label_values).inc(1)
            del g.prometheus_metrics
            return response
