from .server import build_app, run_server

__all__ = ["build_app", "run_server"]
