"""Standalone /metrics sidecar app
(spec: gordo/server/prometheus/server.py:7-25)."""
from flask import Flask
from prometheus_client.exposition import make_wsgi_app
from werkzeug.middleware.dispatcher import DispatcherMiddleware

from .metrics import create_registry


def build_app() -> Flask:
    curr_app = Flask("gordoserver_prometheus")
    registry = create_registry()
    curr_app.wsgi_app = DispatcherMiddleware(
        curr_app.wsgi_app, {"/metrics": make_wsgi_app(registry)}
    )

    @curr_app.route("/healthcheck")
    def health_check():
        return "", 200

    return curr_app
