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
import sys
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


@click.group("workflow")
def workflow_cli():
    """Generate and manage deployment workflows."""


@workflow_cli.command("generate")
@click.option("--machine-config", type=str, required=True,
              envvar=f"{PREFIX}_MACHINE_CONFIG")
@click.option("--project-name", type=str, required=True,
              envvar=f"{PREFIX}_PROJECT_NAME")
@click.option("--project-revision", type=str,
              default="1", envvar=f"{PREFIX}_PROJECT_REVISION")
@click.option("--output-file", type=str, default=None,
              envvar=f"{PREFIX}_OUTPUT_FILE")
@click.option("--docker-registry", type=str, default="ghcr.io",
              envvar=f"{PREFIX}_DOCKER_REGISTRY")
@click.option("--docker-repository", type=str, default="gordo-amd",
              envvar=f"{PREFIX}_DOCKER_REPOSITORY")
@click.option("--gordo-version", type=str, default=gordo_amd.__version__,
              envvar=f"{PREFIX}_GORDO_VERSION")
@click.option("--custom-model-builder-envs", type=str, default=None,
              envvar=f"{PREFIX}_CUSTOM_MODEL_BUILDER_ENVS",
              help="JSON list of extra env vars for builder pods")
@click.option("--split-workflows", type=int, default=30,
              envvar=f"{PREFIX}_SPLIT_WORKFLOWS",
              help="Max machines per rendered workflow document")
@click.option("--n-servers", type=int, default=None,
              envvar=f"{PREFIX}_N_SERVERS")
@click.option("--builder-retries", type=int, default=5,
              envvar=f"{PREFIX}_BUILDER_RETRIES")
@click.option("--keda-enabled", is_flag=True, envvar=f"{PREFIX}_KEDA_ENABLED")
@click.option("--prometheus-server-address", type=str,
              default="http://prometheus:9090",
              envvar=f"{PREFIX}_PROMETHEUS_SERVER_ADDRESS")
@click.option("--default-data-provider", type=str, default=None,
              envvar=f"{PREFIX}_DEFAULT_DATA_PROVIDER")
@click.option("--resource-labels", type=str, default=None,
              envvar=f"{PREFIX}_RESOURCE_LABELS", help="JSON dict of labels")
@click.option("--workflow-template", type=str, default=DEFAULT_TEMPLATE,
              envvar=f"{PREFIX}_WORKFLOW_TEMPLATE")
@click.option("--gpu-fleet", is_flag=True, envvar=f"{PREFIX}_GPU_FLEET",
              help="MI355X mode: one fleet-builder pod (all GPUs) instead "
                   "of one pod per machine")
@click.option("--n-gpus", type=int, default=8, envvar=f"{PREFIX}_N_GPUS")
def workflow_generate(
    machine_config: str,
    project_name: str,
    project_revision: str,
    output_file: Optional[str],
    docker_registry: str,
    docker_repository: str,
    gordo_version: str,
    custom_model_builder_envs: Optional[str],
    split_workflows: int,
    n_servers: Optional[int],
    builder_retries: int,
    keda_enabled: bool,
    prometheus_server_address: str,
    default_data_provider: Optional[str],
    resource_labels: Optional[str],
    workflow_template: str,
    gpu_fleet: bool,
    n_gpus: int,
):
    """Generate the Argo workflow YAML for this config."""
    config = get_dict_from_yaml(machine_config)
    model_builder_env = (
        json.loads(custom_model_builder_envs)
        if custom_model_builder_envs
        else None
    )
    norm = NormalizedConfig(
        config,
        project_name=project_name,
        gordo_version=gordo_version,
        model_builder_env=model_builder_env,
        default_data_provider=default_data_provider,
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

    n_machines = len(norm.machines)
    context = {
        "project_name": project_name,
        "project_revision": project_revision,
        "version": gordo_version,
        "image_pull_policy": default_image_pull_policy(gordo_version),
        "docker_registry": docker_registry,
        "docker_repository": docker_repository,
        "n_servers": n_servers if n_servers is not None else min(n_machines, 10),
        "max_server_replicas": 10 * n_machines,
        "builder_retries": builder_retries,
        "builder_resources": runtime.get("builder", {}).get("resources", {}),
        "server_resources": runtime.get("server", {}).get("resources", {}),
        "builder_env": runtime.get("builder", {}).get("env", []) or [],
        "client_enabled": influx_enabled,
        "client_max_instances": runtime.get("client", {}).get(
            "max_instances", 30
        ),
        "keda_enabled": keda_enabled,
        "prometheus_server_address": prometheus_server_address,
        "resource_labels": json.loads(resource_labels) if resource_labels else None,
        "gpu_fleet_mode": gpu_fleet,
        "n_gpus": n_gpus,
        "service_account": runtime.get("service_account", "gordo-workflow"),
        "models_pvc": runtime.get("models_pvc", "gordo-models"),
        # the server serves every model in the project, independent of how
        # builds were chunked into workflows (reference
        # test_workflow_generator.py::test_expected_models_in_workflow)
        "expected_models_json": json.dumps([m.name for m in norm.machines]),
        # pod/container security contexts from runtime globals (validated
        # by the pydantic schemas via NormalizedConfig.prepare_runtime)
        "pod_security_context": runtime.get("pod_security_context"),
        "security_context": runtime.get("security_context"),
        "prometheus_sidecar": bool(
            runtime.get("prometheus_metrics_server", {}).get("enable", True)
        ),
    }

    template = load_workflow_template(workflow_template)
    documents = []
    chunk = max(1, split_workflows)
    machine_dicts = [
        {"name": m.name, "json": m.to_json()} for m in norm.machines
    ]
    for start in range(0, n_machines, chunk):
        ctx = dict(context)
        ctx["machines"] = machine_dicts[start : start + chunk]
        documents.append(template.render(**ctx))

    output = "\n---\n".join(documents)
    if output_file:
        with open(output_file, "w") as f:
            f.write(output)
    else:
        sys.stdout.write(output)
