"""gunicorn hook cleaning up prometheus multiprocess files on worker
exit (spec: gordo/server/prometheus/gunicorn_config.py)."""
from prometheus_client import multiprocess


def child_exit(server, worker):
    multiprocess.mark_process_dead(worker.pid)
