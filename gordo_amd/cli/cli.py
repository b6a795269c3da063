"""
The ``gordo`` CLI.

Behavioral spec: gordo/cli/cli.py — subcommands ``build`` (env-driven:
MACHINE, OUTPUT_DIR, MODEL_REGISTER_DIR; jinja2 model-parameter
expansion; exception→exit-code table; Katib score printing) and
``run-server``; plus ``workflow`` (in workflow_generator.py) and the
MI355X-native ``fleet`` group (multi-GPU fleet build — no reference
analog; replaces the pod-per-model Argo fan-out on one node).
"""
from __future__ import annotations

import logging
import sys
import traceback
from typing import Any, List, Tuple, cast

import click
import jinja2
import yaml

import gordo_amd
from .custom_types import HostIP, JSONParam, key_value_par
from .exceptions_reporter import ExceptionsReporter, ReportLevel
from .. import serializer
from ..builder.utils import create_model_builder
from ..core.exceptions import (
    ConfigException,
    InsufficientDataError,
    NoSuitableDataProviderError,
    SensorTagNormalizationError,
)
from ..machine import Machine, load_model_config
from ..reporters.exceptions import ReporterException

logger = logging.getLogger(__name__)

_exceptions_reporter = ExceptionsReporter(
    (
        (Exception, 1),
        (ValueError, 2),
        (PermissionError, 20),
        (FileNotFoundError, 30),
        (SensorTagNormalizationError, 60),
        (NoSuitableDataProviderError, 70),
        (InsufficientDataError, 80),
        (ImportError, 85),
        (ReporterException, 90),
        (ConfigException, 100),
    )
)


@click.group("gordo")
@click.version_option(version=gordo_amd.__version__, message=gordo_amd.__version__)
@click.option(
    "--log-level",
    type=str,
    default="INFO",
    help="Run with custom log-level.",
    envvar="GORDO_LOG_LEVEL",
)
@click.pass_context
def gordo(gordo_ctx: click.Context, **ctx):
    """The main entry point for the CLI interface."""
    logging.basicConfig(
        level=getattr(logging, str(gordo_ctx.params.get("log_level")).upper()),
        format=(
            "[%(asctime)s] %(levelname)s "
            "[%(name)s.%(funcName)s:%(lineno)d] %(message)s"
        ),
    )
    gordo_ctx.obj = gordo_ctx.params


@click.command()
@click.argument("machine-config", envvar="MACHINE", type=JSONParam())
@click.argument("output-dir", default="/data", envvar="OUTPUT_DIR")
@click.option(
    "--model-register-dir",
    default=None,
    envvar="MODEL_REGISTER_DIR",
    type=click.Path(exists=False, file_okay=False, dir_okay=True),
)
@click.option(
    "--model-builder-class",
    envvar="MODEL_BUILDER_CLASS",
    type=str,
    default=None,
    help="Import path of a custom ModelBuilder subclass",
)
@click.option(
    "--print-cv-scores",
    help="Prints CV scores to stdout (Katib hyperparameter tuning)",
    is_flag=True,
)
@click.option(
    "--model-parameter",
    type=key_value_par,
    multiple=True,
    default=(),
    help="(key, value) pairs expanded into the jinja2-templated model config",
)
@click.option(
    "--exceptions-reporter-file",
    envvar="EXCEPTIONS_REPORTER_FILE",
    help="JSON output file for exception information",
)
@click.option(
    "--exceptions-report-level",
    type=click.Choice(ReportLevel.get_names(), case_sensitive=False),
    default=ReportLevel.MESSAGE.name,
    envvar="EXCEPTIONS_REPORT_LEVEL",
    help="Details level for exception reporting",
)
def build(
    machine_config: dict,
    output_dir: str,
    model_register_dir,
    model_builder_class: str,
    print_cv_scores: bool,
    model_parameter: List[Tuple[str, Any]],
    exceptions_reporter_file: str,
    exceptions_report_level: str,
):
    """Build a model and deposit it into 'output_dir'."""
    try:
        if model_parameter and isinstance(machine_config.get("model"), str):
            machine_config["model"] = expand_model(
                machine_config["model"], dict(model_parameter)
            )

        machine = Machine.from_config(
            load_model_config(machine_config),
            project_name=machine_config.get("project_name"),
        )

        logger.info("Building, output will be at: %s", output_dir)
        logger.info("Register dir: %s", model_register_dir)

        # canonicalize the model config (all defaults expanded)
        machine.model = serializer.into_definition(
            serializer.from_definition(machine.model)
        )

        cls = create_model_builder(model_builder_class)
        builder = cls(machine=machine)
        _, machine_out = builder.build(output_dir, model_register_dir)

        machine_out.report()

        # NOTE reference fault-injection hook kept for test parity
        # (gordo/cli/cli.py:156-157): machine names containing "err"
        # simulate a builder crash end-to-end.
        if "err" in machine.name:
            raise FileNotFoundError("undefined_file.parquet")

        if print_cv_scores:
            for score in get_all_score_strings(machine_out):
                print(score)
    except Exception:
        traceback.print_exc()
        exc_type, exc_value, exc_traceback = sys.exc_info()
        exit_code = _exceptions_reporter.exception_exit_code(exc_type)
        if exceptions_reporter_file:
            _exceptions_reporter.safe_report(
                cast(
                    ReportLevel,
                    ReportLevel.get_by_name(
                        exceptions_report_level, ReportLevel.EXIT_CODE
                    ),
                ),
                exc_type,
                exc_value,
                exc_traceback,
                exceptions_reporter_file,
                max_message_len=2024 - 500,
            )
        sys.exit(exit_code)
    else:
        return 0


def expand_model(model_config: str, model_parameters: dict) -> dict:
    """Expand a jinja2-templated model config with parameters
    (spec: cli.py:187-216)."""
    try:
        model_template = jinja2.Environment(
            loader=jinja2.BaseLoader(), undefined=jinja2.StrictUndefined
        ).from_string(model_config)
        model_config = model_template.render(**model_parameters)
    except jinja2.exceptions.UndefinedError as e:
        raise ValueError("Model parameter missing value!") from e
    logger.info("Expanded model config: %s", model_config)
    return yaml.safe_load(model_config)


def get_all_score_strings(machine) -> List[str]:
    """'{metric}_{fold}={value}' lines for Katib to scrape
    (spec: cli.py:219-252)."""
    all_scores = []
    for metric_name, scores in (
        machine.metadata.build_metadata.model.cross_validation.scores.items()
    ):
        metric_name = metric_name.replace(" ", "-")
        for score_name, score_val in scores.items():
            score_name = score_name.replace(" ", "-")
            all_scores.append(f"{metric_name}_{score_name}={score_val}")
    return all_scores


@click.command("run-server")
@click.option(
    "--host", type=HostIP(), default="0.0.0.0",
    envvar="GORDO_SERVER_HOST", help="The host to run the server on.",
)
@click.option(
    "--port", type=click.IntRange(1, 65535), default=5555,
    envvar="GORDO_SERVER_PORT", help="The port to run the server on.",
)
@click.option(
    "--workers", type=click.IntRange(1, 4), default=2,
    envvar="GORDO_SERVER_WORKERS", help="The number of worker processes.",
)
@click.option(
    "--worker-connections", type=click.IntRange(1, 4000), default=50,
    envvar="GORDO_SERVER_WORKER_CONNECTIONS",
    help="The maximum number of simultaneous clients per worker process.",
)
@click.option(
    "--threads", type=int, default=8,
    envvar="GORDO_SERVER_THREADS",
    help="Number of request-handling threads per worker (gthread).",
)
@click.option(
    "--worker-class", type=str, default="gthread",
    envvar="GORDO_SERVER_WORKER_CLASS", help="The gunicorn worker class.",
)
@click.option(
    "--log-level",
    type=click.Choice(["debug", "info", "warning", "error", "critical"]),
    default="debug", envvar="GORDO_SERVER_LOG_LEVEL",
    help="The log level for the server.",
)
@click.option(
    "--with-prometheus-config",
    is_flag=True,
    help="Run with the prometheus gunicorn config (multiprocess cleanup)",
)
def run_server_cli(
    host, port, workers, worker_connections, threads, worker_class,
    log_level, with_prometheus_config,
):
    """Run the gordo ML server."""
    from ..server import server

    config_module = None
    if with_prometheus_config:
        config_module = "python:gordo_amd.server.prometheus.gunicorn_config"
    server.run_server(
        host, port, workers, log_level.lower(),
        config_module=config_module,
        worker_connections=worker_connections,
        threads=threads,
        worker_class=worker_class,
    )


@click.command("run-metrics-server")
@click.option("--host", type=HostIP(), default="0.0.0.0",
              envvar="GORDO_METRICS_SERVER_HOST")
@click.option("--port", type=click.IntRange(1, 65535), default=5000,
              envvar="GORDO_METRICS_SERVER_PORT")
def run_metrics_server_cli(host, port):
    """Run the standalone prometheus /metrics sidecar app (the server
    pod's second container — reference template :1147-1180 /
    gordo/server/prometheus/server.py)."""
    from ..server.prometheus.server import build_app

    build_app().run(host=host, port=port)


gordo.add_command(build)
gordo.add_command(run_server_cli)
gordo.add_command(run_metrics_server_cli)

from .workflow_generator import workflow_cli  # noqa: E402

gordo.add_command(workflow_cli)

from .fleet import fleet_cli  # noqa: E402

gordo.add_command(fleet_cli)

from .client import client_cli  # noqa: E402

gordo.add_command(client_cli)

if __name__ == "__main__":
    gordo()
