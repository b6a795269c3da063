"""
``gordo workflow generate`` — YAML machine config → Argo workflow
manifest(s).

Behavioral spec: gordo/cli/workflow_generator.py:132-611 — env-var
backed options (WORKFLOW_GENERATOR_* prefix), NormalizedConfig
construction, context assembly (server replicas = 10×n_machines,
builder/client/server resources, reporter wiring: postgres when influx
enabled, mlflow when runtime.builder.remote_logging.enable), rendering
in chunks of --split-workflows machines separated by '---'.

Adds ``--gpu-fleet`` (MI355X mode): the rendered workflow runs ONE
fleet-builder pod using all node GPUs instead of a pod per machine.
"""
from __future__ import annotations

import json
import logging
import os
import re
import sys
import time
from typing import Optional

import click

import gordo_amd
from ..workflow.config_elements.normalized_config import NormalizedConfig
from ..workflow.workflow_generator import (
    default_image_pull_policy,
    get_dict_from_yaml,
    load_workflow_template,
)

logger = logging.getLogger(__name__)

PREFIX = "WORKFLOW_GENERATOR"
DEFAULT_TEMPLATE = os.path.join(
    os.path.dirname(os.path.dirname(__file__)),
    "workflow",
    "workflow_generator",
    "resources",
    "argo-workflow.yml.template",
)

# HPA flavors for the ML server (reference workflow_generator.py:37-38)
ML_SERVER_HPA_TYPES = ["none", "k8s_cpu", "keda"]
DEFAULT_ML_SERVER_HPA_TYPE = "k8s_cpu"

# KEDA prometheus-scaler defaults (reference workflow_generator.py:40-42)
DEFAULT_KEDA_PROMETHEUS_METRIC_NAME = (
    "gordo_server_request_duration_seconds_count"
)
DEFAULT_KEDA_PROMETHEUS_QUERY = (
    'sum(rate(gordo_server_request_duration_seconds_count'
    '{project=~"{{project_name}}",path=~".*prediction"}[30s]))'
)
DEFAULT_KEDA_PROMETHEUS_THRESHOLD = "1.0"


def _valid_owner_ref(ctx, param, value):
    """Owner references: nonempty JSON/YAML list of dicts each carrying
    at least uid/name/kind/apiVersion (reference wg._valid_owner_ref)."""
    if value is None or value == "":
        return None
    try:
        parsed = json.loads(value)
    except json.JSONDecodeError:
        import yaml

        parsed = yaml.safe_load(value)
    if not isinstance(parsed, list) or not parsed:
        raise click.BadParameter(
            "owner-references must be a nonempty list of owner-reference "
            "dicts"
        )
    for ref in parsed:
        missing = {"uid", "name", "kind", "apiVersion"} - set(ref)
        if missing:
            raise click.BadParameter(
                f"owner-reference missing keys: {sorted(missing)}"
            )
    return parsed


def _labels_param(value: Optional[str], name: str) -> Optional[dict]:
    """'' or a JSON dict (reference --resources-labels contract)."""
    if not value:
        return None
    parsed = json.loads(value)
    if not isinstance(parsed, dict):
        raise click.BadParameter(f"{name} must be a JSON dictionary")
    return parsed


@click.group("workflow")
def workflow_cli():
    """Generate and manage deployment workflows."""


@workflow_cli.command("generate")
@click.option("--machine-config", type=str, required=True,
              envvar=f"{PREFIX}_MACHINE_CONFIG",
              help="Machine configuration file")
@click.option("--workflow-template", type=str, default=DEFAULT_TEMPLATE,
              envvar=f"{PREFIX}_WORKFLOW_TEMPLATE",
              help="Template to expand")
@click.option("--owner-references", callback=_valid_owner_ref, default=None,
              envvar=f"{PREFIX}_OWNER_REFERENCES",
              help="K8s owner references injected into all created "
                   "resources: nonempty yaml/json list of dicts with at "
                   "least uid, name, kind, apiVersion")
@click.option("--gordo-version", type=str, default=gordo_amd.__version__,
              envvar=f"{PREFIX}_GORDO_VERSION",
              help="Version of gordo to use, if different than this one")
@click.option("--project-name", type=str, required=True,
              envvar=f"{PREFIX}_PROJECT_NAME",
              help="Name of the project which owns the workflow")
@click.option("--project-revision", type=str,
              default=lambda: str(int(time.time() * 1000)),
              envvar=f"{PREFIX}_PROJECT_REVISION",
              help="Revision of the project (default: unix time ms)")
@click.option("--output-file", type=str, default=None,
              envvar=f"{PREFIX}_OUTPUT_FILE")
@click.option("--namespace", type=str, default="kubeflow",
              envvar=f"{PREFIX}_NAMESPACE",
              help="Which namespace to deploy services into")
@click.option("--split-workflows", type=int, default=30,
              envvar=f"{PREFIX}_SPLIT_WORKFLOWS",
              help="Max machines per rendered workflow document")
@click.option("--n-servers", type=int, default=None,
              envvar=f"{PREFIX}_N_SERVERS",
              help="Max number of ML servers, default n_machines*10")
@click.option("--docker-repository", type=str, default="gordo-amd",
              envvar=f"{PREFIX}_DOCKER_REPOSITORY")
@click.option("--docker-registry", type=str, default="ghcr.io",
              envvar=f"{PREFIX}_DOCKER_REGISTRY")
@click.option("--retry-backoff-duration", type=str, default="15s",
              envvar=f"{PREFIX}_RETRY_BACKOFF_DURATION",
              help="retryStrategy.backoff.duration for workflow steps")
@click.option("--retry-backoff-factor", type=int, default=2,
              envvar=f"{PREFIX}_RETRY_BACKOFF_FACTOR",
              help="retryStrategy.backoff.factor for workflow steps")
@click.option("--builder-retries", type=int, default=5,
              envvar=f"{PREFIX}_BUILDER_RETRIES",
              help="retryStrategy.limit for workflow steps")
@click.option("--gordo-server-workers", type=int, default=None,
              envvar=f"{PREFIX}_GORDO_SERVER_WORKERS",
              help="Number of server worker processes")
@click.option("--gordo-server-threads", type=int, default=None,
              envvar=f"{PREFIX}_GORDO_SERVER_THREADS",
              help="Number of server worker threads")
@click.option("--gordo-server-probe-timeout", type=int, default=None,
              envvar=f"{PREFIX}_GORDO_SERVER_PROBE_TIMEOUT",
              help="timeoutSeconds for server liveness/readiness probes")
@click.option("--without-prometheus", is_flag=True,
              envvar=f"{PREFIX}_WITHOUT_PROMETHEUS",
              help="Do not deploy the prometheus metrics sidecar")
@click.option("--prometheus-metrics-server-workers", type=int, default=1,
              envvar=f"{PREFIX}_PROMETHEUS_METRICS_SERVER_WORKERS")
@click.option("--image-pull-policy", type=str, default=None,
              envvar=f"{PREFIX}_IMAGE_PULL_POLICY",
              help="Default imagePullPolicy for all gordo images")
@click.option("--with-keda", "--keda-enabled", "with_keda", is_flag=True,
              envvar=f"{PREFIX}_WITH_KEDA",
              help="Enable support for the KEDA autoscaler")
@click.option("--ml-server-hpa-type",
              type=click.Choice(ML_SERVER_HPA_TYPES),
              default=DEFAULT_ML_SERVER_HPA_TYPE,
              envvar=f"{PREFIX}_ML_SERVER_HPA_TYPE",
              help="HPA type for the ML server")
@click.option("--custom-model-builder-envs", type=str, default="[]",
              envvar=f"{PREFIX}_CUSTOM_MODEL_BUILDER_ENVS",
              help="JSON list of extra env vars for builder pods")
@click.option("--prometheus-server-address", type=str, default=None,
              envvar=f"{PREFIX}_PROMETHEUS_SERVER_ADDRESS",
              help='Prometheus url, required for --ml-server-hpa-type=keda')
@click.option("--keda-prometheus-metric-name", type=str,
              default=DEFAULT_KEDA_PROMETHEUS_METRIC_NAME,
              envvar=f"{PREFIX}_KEDA_PROMETHEUS_METRIC_NAME")
@click.option("--keda-prometheus-query", type=str,
              default=DEFAULT_KEDA_PROMETHEUS_QUERY,
              envvar=f"{PREFIX}_KEDA_PROMETHEUS_QUERY")
@click.option("--keda-prometheus-threshold", type=str,
              default=DEFAULT_KEDA_PROMETHEUS_THRESHOLD,
              envvar=f"{PREFIX}_KEDA_PROMETHEUS_THRESHOLD")
@click.option("--resources-labels", "--resource-labels", "resources_labels",
              type=str, default="",
              envvar=f"{PREFIX}_RESOURCE_LABELS",
              help="Additional labels for resources ('' or a JSON dict)")
@click.option("--model-builder-labels", type=str, default="",
              envvar=f"{PREFIX}_MODEL_BUILDER_LABELS",
              help="Additional labels for the model-builder step "
                   "('' or a JSON dict)")
@click.option("--server-labels", type=str, default="",
              envvar=f"{PREFIX}_SERVER_LABELS",
              help="Additional labels for gordo-server ('' or a JSON dict)")
@click.option("--server-termination-grace-period", type=int, default=60,
              envvar=f"{PREFIX}_SERVER_TERMINATION_GRACE_PERIOD")
@click.option("--server-target-cpu-utilization-percentage", type=int,
              default=50,
              envvar=f"{PREFIX}_SERVER_TARGET_CPU_UTILIZATION_PERCENTAGE")
@click.option("--gordo-server-readiness-initial-delay", type=int, default=5,
              envvar=f"{PREFIX}_GORDO_SERVER_READINESS_INITIAL_DELAY")
@click.option("--gordo-server-liveness-initial-delay", type=int, default=600,
              envvar=f"{PREFIX}_GORDO_SERVER_LIVENESS_INITIAL_DELAY")
@click.option("--security-context", type=str, default=None,
              envvar=f"{PREFIX}_SECURITY_CONTEXT",
              help="Containers securityContext in JSON format")
@click.option("--pod-security-context", type=str, default=None,
              envvar=f"{PREFIX}_POD_SECURITY_CONTEXT",
              help="Global workflow securityContext in JSON format")
@click.option("--default-data-provider", type=str, default=None,
              envvar=f"{PREFIX}_DEFAULT_DATA_PROVIDER",
              help="Default data_provider.type for dataset")
@click.option("--model-builder-class", type=str, default=None,
              envvar="MODEL_BUILDER_CLASS", help="ModelBuilder class")
@click.option("--argo-binary", type=str, default="argo",
              help="Argo binary path (argo, argo2, ...)")
@click.option("--gpu-fleet", is_flag=True, envvar=f"{PREFIX}_GPU_FLEET",
              help="MI355X mode: one fleet-builder pod (all GPUs) instead "
                   "of one pod per machine")
@click.option("--n-gpus", type=int, default=8, envvar=f"{PREFIX}_N_GPUS")
def workflow_generate(**opts):
    """Generate the Argo workflow YAML for this config."""
    if not re.match(r"^argo\d*$", opts["argo_binary"]):
        raise click.BadParameter("--argo-binary must match ^argo\\d*$")
    if (
        opts["ml_server_hpa_type"] == "keda"
        and not opts["with_keda"]
    ):
        raise click.ClickException(
            '"--with-keda" is required for --ml-server-hpa-type=keda'
        )
    if (
        opts["ml_server_hpa_type"] == "keda"
        and not opts["prometheus_server_address"]
    ):
        raise click.ClickException(
            '"--prometheus-server-address" is required for '
            '--ml-server-hpa-type=keda'
        )

    config = get_dict_from_yaml(opts["machine_config"])
    model_builder_env = (
        json.loads(opts["custom_model_builder_envs"])
        if opts["custom_model_builder_envs"]
        else None
    )
    project_name = opts["project_name"]
    gordo_version = opts["gordo_version"]
    norm = NormalizedConfig(
        config,
        project_name=project_name,
        gordo_version=gordo_version,
        model_builder_env=model_builder_env,
        default_data_provider=opts["default_data_provider"],
    )

    # wire reporters: postgres when influx enabled; mlflow when
    # runtime.builder.remote_logging.enable
    runtime = norm.globals.get("runtime", {})
    influx_enabled = runtime.get("influx", {}).get("enable", False)
    remote_logging = (
        runtime.get("builder", {}).get("remote_logging", {}).get("enable", False)
    )
    for machine in norm.machines:
        reporters = machine.runtime.setdefault("reporters", [])
        if influx_enabled:
            reporters.append(
                {
                    "gordo_amd.reporters.postgres.PostgresReporter": {
                        "host": f"gordo-postgres-{project_name}"
                    }
                }
            )
        if remote_logging:
            reporters.append(
                {"gordo_amd.reporters.mlflow.MlFlowReporter": {}}
            )

    security_context = (
        json.loads(opts["security_context"])
        if opts["security_context"]
        else runtime.get("security_context")
    )
    pod_security_context = (
        json.loads(opts["pod_security_context"])
        if opts["pod_security_context"]
        else runtime.get("pod_security_context")
    )

    n_machines = len(norm.machines)
    n_servers = opts["n_servers"]
    context = {
        "project_name": project_name,
        "project_revision": opts["project_revision"],
        "version": gordo_version,
        "namespace": opts["namespace"],
        "image_pull_policy": (
            opts["image_pull_policy"]
            or default_image_pull_policy(gordo_version)
        ),
        "docker_registry": opts["docker_registry"],
        "docker_repository": opts["docker_repository"],
        "n_servers": n_servers if n_servers is not None else min(n_machines, 10),
        "max_server_replicas": 10 * n_machines,
        "builder_retries": opts["builder_retries"],
        "retry_backoff_duration": opts["retry_backoff_duration"],
        "retry_backoff_factor": opts["retry_backoff_factor"],
        "builder_resources": runtime.get("builder", {}).get("resources", {}),
        "server_resources": runtime.get("server", {}).get("resources", {}),
        "influx_resources": runtime.get("influx", {}).get("resources", {}),
        "builder_env": runtime.get("builder", {}).get("env", []) or [],
        "influx_enabled": influx_enabled,
        "client_enabled": influx_enabled,
        "client_max_instances": runtime.get("client", {}).get(
            "max_instances", 30
        ),
        "ml_server_hpa_type": opts["ml_server_hpa_type"],
        "with_keda": opts["with_keda"],
        "prometheus_server_address": opts["prometheus_server_address"],
        "keda_prometheus_metric_name": opts["keda_prometheus_metric_name"],
        "keda_prometheus_query": opts["keda_prometheus_query"].replace(
            "{{project_name}}", project_name
        ),
        "keda_prometheus_threshold": opts["keda_prometheus_threshold"],
        "server_target_cpu_utilization_percentage": opts[
            "server_target_cpu_utilization_percentage"
        ],
        "server_termination_grace_period": opts[
            "server_termination_grace_period"
        ],
        "server_readiness_initial_delay": opts[
            "gordo_server_readiness_initial_delay"
        ],
        "server_liveness_initial_delay": opts[
            "gordo_server_liveness_initial_delay"
        ],
        "server_probe_timeout": opts["gordo_server_probe_timeout"],
        "gordo_server_workers": opts["gordo_server_workers"],
        "gordo_server_threads": opts["gordo_server_threads"],
        "resource_labels": _labels_param(
            opts["resources_labels"], "resources-labels"
        ),
        "model_builder_labels": _labels_param(
            opts["model_builder_labels"], "model-builder-labels"
        ),
        "server_labels": _labels_param(opts["server_labels"], "server-labels"),
        "owner_references_json": (
            json.dumps(opts["owner_references"])
            if opts["owner_references"]
            else None
        ),
        "model_builder_class": opts["model_builder_class"],
        "argo_binary": opts["argo_binary"],
        "gpu_fleet_mode": opts["gpu_fleet"],
        "n_gpus": opts["n_gpus"],
        "service_account": runtime.get("service_account", "gordo-workflow"),
        "models_pvc": runtime.get("models_pvc", "gordo-models"),
        # the server serves every model in the project, independent of how
        # builds were chunked into workflows (reference
        # test_workflow_generator.py::test_expected_models_in_workflow)
        "expected_models_json": json.dumps([m.name for m in norm.machines]),
        # pod/container security contexts from CLI JSON or runtime globals
        # (validated by the pydantic schemas via prepare_runtime)
        "pod_security_context": pod_security_context,
        "security_context": security_context,
        "prometheus_sidecar": (
            not opts["without_prometheus"]
            and bool(
                runtime.get("prometheus_metrics_server", {}).get(
                    "enable", True
                )
            )
        ),
        "prometheus_metrics_server_workers": opts[
            "prometheus_metrics_server_workers"
        ],
    }

    template = load_workflow_template(opts["workflow_template"])
    documents = []
    chunk = max(1, opts["split_workflows"])
    machine_dicts = [
        {"name": m.name, "json": m.to_json()} for m in norm.machines
    ]
    for start in range(0, n_machines, chunk):
        ctx = dict(context)
        ctx["machines"] = machine_dicts[start : start + chunk]
        documents.append(template.render(**ctx))

    output = "\n---\n".join(documents)
    if opts["output_file"]:
        with open(opts["output_file"], "w") as f:
            f.write(output)
    else:
        sys.stdout.write(output)
