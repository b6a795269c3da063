"""
NormalizedConfig — default globals + per-machine patching.

Behavioral spec: gordo/workflow/config_elements/normalized_config.py —
default runtime resources & evaluation globals (cv_mode full_build,
MinMaxScaler scoring scaler, the 4 default metrics), influx resources
scaled with machine count, globals merged into each machine via
``patch_dict``, runtime validated through the pydantic schemas.
"""
from __future__ import annotations

from copy import deepcopy
from typing import Any, Dict, List, Optional, Type

from pydantic import TypeAdapter

from .schemas import (
    BuilderPodRuntime,
    PodRuntime,
    PodSecurityContext,
    SecurityContext,
    Volume,
)
from ..workflow_generator.helpers import patch_dict
from ...machine import Machine, load_globals_config, load_machine_config
from ...machine.validators import fix_runtime
from ...utils import join_json_paths

import gordo_amd


def _calculate_influx_resources(nr_of_machines: int) -> Dict[str, Any]:
    return {
        "requests": {
            "memory": min(3000 + (220 * nr_of_machines), 28000),
            "cpu": min(500 + (10 * nr_of_machines), 4000),
        },
        "limits": {
            "memory": min(3000 + (220 * nr_of_machines), 48000),
            "cpu": 10000 + (20 * nr_of_machines),
        },
    }


class NormalizedConfig:
    """YAML config dict (machines + globals) → list of validated
    ``Machine`` objects with all defaults applied."""

    UNIFIED_DOCKER_IMAGES: Dict[str, Any] = {
        "runtime": {
            "deployer": {"image": "gordo-base"},
            "server": {"image": "gordo-base"},
            "prometheus_metrics_server": {"image": "gordo-base"},
            "builder": {"image": "gordo-base"},
            "client": {"image": "gordo-base"},
        }
    }

    DEFAULT_CONFIG_GLOBALS: Dict[str, Any] = {
        "runtime": {
            "reporters": [],
            "server": {
                "resources": {
                    "requests": {"memory": 3000, "cpu": 1000},
                    "limits": {"memory": 6000, "cpu": 2000},
                }
            },
            "prometheus_metrics_server": {
                "resources": {
                    "requests": {"memory": 200, "cpu": 100},
                    "limits": {"memory": 1000, "cpu": 200},
                }
            },
            "builder": {
                "resources": {
                    "requests": {"memory": 3900, "cpu": 1001},
                    "limits": {"memory": 31200, "cpu": 1001},
                },
                "remote_logging": {"enable": False},
            },
            "client": {
                "resources": {
                    "requests": {"memory": 3500, "cpu": 100},
                    "limits": {"memory": 4000, "cpu": 2000},
                },
                "max_instances": 30,
            },
            "influx": {"enable": True},
        },
        "evaluation": {
            "cv_mode": "full_build",
            "scoring_scaler": "sklearn.preprocessing.MinMaxScaler",
            "metrics": [
                "explained_variance_score",
                "r2_score",
                "mean_squared_error",
                "mean_absolute_error",
            ],
        },
    }

    def __init__(
        self,
        config: dict,
        project_name: str,
        gordo_version: Optional[str] = None,
        model_builder_env: Optional[dict] = None,
        default_data_provider: Optional[str] = None,
        json_path: Optional[str] = None,
    ):
        if gordo_version is None:
            gordo_version = gordo_amd.__version__
        default_globals = self.get_default_globals(gordo_version)
        default_globals["runtime"]["influx"]["resources"] = (
            _calculate_influx_resources(len(config["machines"]))
        )

        passed_globals = load_globals_config(
            config.get("globals", {}), join_json_paths("globals", json_path or "")
        )

        if model_builder_env is not None:
            builder = default_globals.setdefault("runtime", {}).setdefault(
                "builder", {}
            )
            builder.setdefault("env", model_builder_env)

        patched_globals = patch_dict(default_globals, passed_globals)
        patched_globals = self.prepare_patched_globals(patched_globals)

        self.project_name = project_name
        self.machines: List[Machine] = []
        for i, conf in enumerate(config["machines"]):
            machine_config = load_machine_config(
                conf, join_json_paths(f"machines[{i}]", json_path or "")
            )
            self.machines.append(
                Machine.from_config(
                    machine_config,
                    project_name=project_name,
                    config_globals=patched_globals,
                    default_data_provider=default_data_provider,
                )
            )
        self.globals: Dict[str, Any] = patched_globals

    @staticmethod
    def prepare_runtime(runtime: dict) -> dict:
        def prepare_pod_runtime(name: str, schema: Type[PodRuntime] = PodRuntime):
            if name in runtime:
                pod_runtime = TypeAdapter(schema).validate_python(runtime[name])
                runtime[name] = pod_runtime.model_dump(exclude_none=True)

        prepare_pod_runtime("builder", BuilderPodRuntime)
        if "pod_security_context" in runtime:
            psc = TypeAdapter(PodSecurityContext).validate_python(
                runtime["pod_security_context"]
            )
            runtime["pod_security_context"] = psc.model_dump(exclude_none=True)
        if "security_context" in runtime:
            sc = TypeAdapter(SecurityContext).validate_python(
                runtime["security_context"]
            )
            runtime["security_context"] = sc.model_dump(exclude_none=True)
        if "volumes" in runtime:
            volumes = TypeAdapter(List[Volume]).validate_python(runtime["volumes"])
            runtime["volumes"] = [
                v.model_dump(exclude_none=True) for v in volumes
            ]
        return runtime

    @classmethod
    def prepare_patched_globals(cls, patched_globals: dict) -> dict:
        runtime = fix_runtime(patched_globals.get("runtime"))
        runtime = cls.prepare_runtime(runtime)
        patched_globals["runtime"] = runtime
        return patched_globals

    @classmethod
    def get_default_globals(cls, gordo_version: str) -> dict:
        return patch_dict(
            deepcopy(cls.DEFAULT_CONFIG_GLOBALS), cls.UNIFIED_DOCKER_IMAGES
        )
