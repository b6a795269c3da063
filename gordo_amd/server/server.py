"""
Flask app factory + server launcher.

Behavioral spec: gordo/server/server.py — env-driven Config
(MODEL_COLLECTION_DIR, EXPECTED_MODELS, ENABLE_PROMETHEUS, PROJECT),
base+anomaly blueprints, Envoy/Ambassador prefix-rewriting middleware,
per-request revision resolution with 410 on missing, response stamping
(revision + Server-Timing), /healthcheck and /server-version.

``run_server`` launches gunicorn when available (the reference's
worker model) and falls back to werkzeug's threaded server in
environments without gunicorn (this image).
"""
from __future__ import annotations

import functools
import json
import logging
import os
import subprocess
import timeit
import typing
from typing import Any, Dict, Optional

import yaml
from flask import Flask, current_app, g, jsonify, make_response, request

import gordo_amd
from . import blueprints
from .utils import validate_revision

logger = logging.getLogger(__name__)


def enable_prometheus() -> bool:
    return os.getenv("ENABLE_PROMETHEUS", "false") != "false"


class Config:
    """Server config, read from the environment."""

    def __init__(self):
        self.MODEL_COLLECTION_DIR_ENV_VAR = "MODEL_COLLECTION_DIR"
        self.EXPECTED_MODELS = yaml.safe_load(os.getenv("EXPECTED_MODELS", "[]"))
        self.ENABLE_PROMETHEUS = enable_prometheus()
        self.PROJECT = os.getenv("PROJECT")


def adapt_proxy_deployment(wsgi_app: typing.Callable) -> typing.Callable:
    """WSGI wrapper mapping Envoy/Ambassador-prefixed paths
    (HTTP_X_ENVOY_ORIGINAL_PATH) onto the locally-known routes
    (spec: gordo/server/server.py:46-118).

    >>> app = Flask(__name__)
    >>> app.wsgi_app = adapt_proxy_deployment(app.wsgi_app)
    """

    @functools.wraps(wsgi_app)
    def wrapper(environ, start_response):
        script_name = environ.get("HTTP_X_ENVOY_ORIGINAL_PATH", "")
        if script_name:
            path_info = environ.get("PATH_INFO", "")
            if path_info.rstrip("/"):
                script_name = script_name.replace(path_info, "")
            environ["SCRIPT_NAME"] = script_name
            if path_info.startswith(script_name):
                environ["PATH_INFO"] = path_info[len(script_name):]
        scheme = environ.get("HTTP_X_FORWARDED_PROTO", "")
        if scheme:
            environ["wsgi.url_scheme"] = scheme
        return wsgi_app(environ, start_response)

    return wrapper


def build_app(
    config: Optional[Dict[str, Any]] = None,
    prometheus_registry=None,
) -> Flask:
    """Build the serving app with all routes registered."""
    app = Flask(__name__)
    app.config.from_object(Config())
    if config is not None:
        app.config.update(**config)

    app.register_blueprint(blueprints.base_blueprint)
    app.register_blueprint(blueprints.anomaly_blueprint)

    app.wsgi_app = adapt_proxy_deployment(app.wsgi_app)  # type: ignore
    app.url_map.strict_slashes = False

    if app.config["ENABLE_PROMETHEUS"]:
        from .prometheus.metrics import GordoServerPrometheusMetrics

        prometheus_metrics = GordoServerPrometheusMetrics(
            args_labels=[("gordo_project", "project"), ("gordo_name", "model")],
            info={"version": gordo_amd.__version__},
            # probes shouldn't pollute the request metrics (reference
            # tests/gordo/server/test_prometheus.py::test_ignore)
            ignore_paths=["/healthcheck"],
            registry=prometheus_registry,
        )
        prometheus_metrics.prepare_app(app)
    elif prometheus_registry is not None:
        logger.warning("Ignoring non empty prometheus_registry argument")

    @app.before_request
    def _start_timer():
        g.start_time = timeit.default_timer()

    @app.before_request
    def _set_revision_and_collection_dir():
        g.collection_dir = os.environ[
            current_app.config["MODEL_COLLECTION_DIR_ENV_VAR"]
        ]
        g.current_revision = os.path.basename(g.collection_dir)

        has_revision = (
            "revision" in request.args or "revision" in request.headers
        )
        if has_revision:
            g.revision = request.args.get(
                "revision", request.headers.get("revision")
            )
            if not validate_revision(g.revision):
                return make_response(
                    jsonify({"error": "Revision should only contains numbers."}),
                    410,
                )
            g.collection_dir = os.path.join(g.collection_dir, "..", g.revision)
            try:
                os.listdir(g.collection_dir)
            except FileNotFoundError:
                return make_response(
                    jsonify({"error": f"Revision '{g.revision}' not found."}), 410
                )
        else:
            g.revision = g.current_revision

    @app.after_request
    def _revision_used(response):
        if response.is_json:
            data = response.get_json()
            if data is not None:
                data["revision"] = g.revision
                response.set_data(json.dumps(data).encode())
        response.headers["revision"] = g.revision
        return response

    @app.after_request
    def _log_time_taken(response):
        runtime_s = timeit.default_timer() - g.start_time
        logger.debug("Total runtime for request: %ss", runtime_s)
        response.headers["Server-Timing"] = f"request_walltime_s;dur={runtime_s}"
        return response

    @app.route("/healthcheck")
    def base_healthcheck():
        return "", 200

    @app.route("/server-version")
    def server_version():
        return jsonify({"version": gordo_amd.__version__})

    return app


def run_cmd(cmd):
    """Run a shell command, sending stderr to stdout."""
    subprocess.check_call(cmd, stderr=subprocess.STDOUT)


def run_server(
    host: str,
    port: int,
    workers: int,
    log_level: str,
    config_module: Optional[str] = None,
    worker_connections: Optional[int] = None,
    threads: Optional[int] = None,
    worker_class: str = "gthread",
    server_app: str = "gordo_amd.server.server:build_app()",
):
    """Launch the model server: gunicorn when importable (reference
    worker model, server.py:240-304), else werkzeug."""
    try:
        import gunicorn  # noqa: F401

        has_gunicorn = True
    except ImportError:
        has_gunicorn = False

    if has_gunicorn:
        cmd = [
            "gunicorn",
            "--bind", f"{host}:{port}",
            "--log-level", log_level,
            "--error-logfile", "-",
            "--access-logfile", "-",
            "--worker-tmp-dir", "/dev/shm",
            "--worker-class", worker_class,
            "--workers", str(workers),
        ]
        if worker_class == "gthread" and threads is not None:
            cmd += ["--threads", str(threads)]
        if worker_class == "gevent" and worker_connections is not None:
            cmd += ["--worker-connections", str(worker_connections)]
        if config_module:
            cmd += ["--config", config_module]
        cmd.append(server_app)
        run_cmd(cmd)
    elif workers and workers > 1:
        logger.warning(
            "gunicorn is not installed; serving with a prefork werkzeug "
            "pool (%d worker processes x threads)", workers,
        )
        _run_prefork(host, port, workers)
    else:
        logger.warning(
            "gunicorn is not installed; serving with werkzeug (threaded)"
        )
        app = build_app()
        app.run(host=host, port=port, threaded=True, debug=False)


def _run_prefork(host: str, port: int, workers: int):
    """Multi-process serving without gunicorn: one shared listening
    socket, N worker processes accepting from it, parent supervises
    and restarts dead workers (the reference's gunicorn master/worker
    model, server.py:240-304 — worker processes are what scale the
    GIL-bound request path; each worker holds its own model LRU exactly
    like a gunicorn worker would).

    Workers are EXEC'd (fresh interpreters inheriting the listener fd),
    not forked: torch poisons CUDA in forked children once the parent
    has touched the CUDA runtime ("Cannot re-initialize CUDA in forked
    subprocess" — observed on the round-2 GPU lease), and exec'd
    workers each own a clean HIP context.
    """
    import signal
    import socket
    import subprocess
    import sys
    import time as _time

    sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    sock.bind((host, port))
    sock.listen(1024)
    sock.set_inheritable(True)
    fd = sock.fileno()

    def spawn():
        env = dict(os.environ)
        env["GORDO_SERVER_FD"] = str(fd)
        env["GORDO_SERVER_HOST"] = host
        env["GORDO_SERVER_PORT"] = str(port)
        return subprocess.Popen(
            [sys.executable, "-m", "gordo_amd.server.worker"],
            env=env,
            pass_fds=(fd,),
        )

    procs = [spawn() for _ in range(workers)]
    stopping = {"flag": False}

    def on_term(signum, frame):
        stopping["flag"] = True
        for p in procs:
            if p.poll() is None:
                p.terminate()

    signal.signal(signal.SIGTERM, on_term)
    signal.signal(signal.SIGINT, on_term)
    try:
        while True:
            if stopping["flag"]:
                for p in procs:
                    try:
                        p.wait(timeout=20)
                    except subprocess.TimeoutExpired:
                        p.kill()
                return
            for i, p in enumerate(procs):
                rc = p.poll()
                if rc is not None and not stopping["flag"]:
                    logger.warning(
                        "serving worker %d exited rc=%s; restarting",
                        p.pid, rc,
                    )
                    procs[i] = spawn()
            _time.sleep(0.5)
    finally:
        sock.close()
