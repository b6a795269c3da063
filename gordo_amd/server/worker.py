"""Serving worker: one exec'd process of the prefork pool
(server._run_prefork). Accepts on the listener fd inherited from the
master and serves the Flask app with werkzeug threads — the
gunicorn-worker analog (reference gordo/server/server.py:240-304),
with its own HIP context and model LRU."""
from __future__ import annotations

import logging
import os


def main():
    logging.basicConfig(
        level=os.environ.get("GORDO_LOG_LEVEL", "INFO").upper()
    )
    fd = int(os.environ["GORDO_SERVER_FD"])
    host = os.environ.get("GORDO_SERVER_HOST", "0.0.0.0")
    port = int(os.environ.get("GORDO_SERVER_PORT", "5555"))
    from werkzeug.serving import make_server

    from .server import build_app

    srv = make_server(host, port, build_app(), threaded=True, fd=fd)
    srv.serve_forever()


if __name__ == "__main__":
    main()
