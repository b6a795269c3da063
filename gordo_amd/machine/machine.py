"""
Machine — the per-asset unit of configuration.

Behavioral spec (gordo/machine/machine.py:30-269): validated attributes
via descriptors; ``from_config`` merges globals via ``patch_dict`` with
the reference's exact precedence — runtime/evaluation: machine overlays
globals; dataset: GLOBALS overlay the machine's dataset (reference
machine.py:123-125); model: machine's model, else globals'.
YAML/JSON round-trips with ``|``-multiline nested fields.
"""
from __future__ import annotations

import json
from typing import Any, Dict, List, Optional

import yaml

from .encoders import MachineJSONEncoder, MachineSafeDumper
from .metadata import Metadata
from .validators import (
    ValidDataset,
    ValidMachineRuntime,
    ValidMetadata,
    ValidModel,
    ValidUrlString,
)
from ..core.base import GordoBaseDataset
from ..core.sensor_tag import SensorTag


class Machine:
    name = ValidUrlString()
    project_name = ValidUrlString()
    host = ValidUrlString()
    model = ValidModel()
    dataset = ValidDataset()
    metadata = ValidMetadata()
    runtime = ValidMachineRuntime()

    @staticmethod
    def prepare_evaluation(evaluation: Optional[dict]) -> dict:
        if evaluation is None:
            return dict(cv_mode="full_build")
        return evaluation

    def __init__(
        self,
        name: str,
        model: dict,
        dataset: GordoBaseDataset,
        project_name: str,
        evaluation: Optional[dict] = None,
        metadata: Optional[Metadata] = None,
        runtime: Optional[dict] = None,
    ):
        if runtime is None:
            runtime = {}
        if metadata is None:
            metadata = Metadata.from_dict({})
        if isinstance(metadata, dict):
            metadata = Metadata.from_dict(metadata)
        if isinstance(dataset, dict):
            dataset = GordoBaseDataset.from_dict(dataset)
        self.name = name
        self.model = model
        self.dataset = dataset
        self.runtime = runtime
        self.evaluation = self.prepare_evaluation(evaluation)
        self.metadata = metadata
        self.project_name = project_name
        self.host = f"gordoserver-{self.project_name}-{self.name}"

    @classmethod
    def from_config(
        cls,
        config: Dict[str, Any],
        project_name: Optional[str] = None,
        config_globals: Optional[Dict[str, Any]] = None,
        default_data_provider: Optional[str] = None,
    ) -> "Machine":
        from ..workflow.workflow_generator.helpers import patch_dict

        if config_globals is None:
            config_globals = {}

        name = config["name"]
        model = config.get("model") or config_globals.get("model")
        if model is None:
            raise ValueError("model is empty")

        if project_name is None:
            project_name = config.get("project_name")
        if project_name is None:
            raise ValueError("project_name is empty")

        runtime = patch_dict(
            config_globals.get("runtime", {}), config.get("runtime", {})
        )
        # NOTE reference quirk kept for compatibility: globals overlay the
        # machine's dataset (machine.py:123-125), unlike model/runtime.
        dataset = patch_dict(
            config.get("dataset", {}), config_globals.get("dataset", {})
        )
        evaluation = patch_dict(
            config_globals.get("evaluation", {}),
            cls.prepare_evaluation(config.get("evaluation")),
        )
        if default_data_provider and "data_provider" not in dataset:
            dataset["data_provider"] = {"type": default_data_provider}

        metadata = Metadata(
            user_defined={
                "global-metadata": config_globals.get("metadata", {}),
                "machine-metadata": config.get("metadata", {}),
            }
        )
        return cls.from_dict(
            {
                "name": name,
                "model": model,
                "dataset": dataset,
                "project_name": project_name,
                "evaluation": evaluation,
                "metadata": metadata,
                "runtime": runtime,
            }
        )

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Machine":
        metadata = d.get("metadata")
        if isinstance(metadata, dict):
            metadata = Metadata.from_dict(metadata)
        dataset = d["dataset"]
        if isinstance(dataset, dict):
            dataset = GordoBaseDataset.from_dict(dataset)
        return cls(
            name=d["name"],
            model=d["model"],
            dataset=dataset,
            project_name=d["project_name"],
            evaluation=d.get("evaluation"),
            metadata=metadata,
            runtime=d.get("runtime"),
        )

    def normalize_sensor_tags(self, tag_list: List) -> List[SensorTag]:
        """Resolve tag metadata for a list of tags using the dataset's
        build metadata (spec: gordo/machine/machine.py:151)."""
        from ..utils import normalize_sensor_tags

        metadata = self.metadata.to_dict() if self.metadata else {}
        build_dataset_metadata = (
            metadata.get("build_metadata", {}).get("dataset", {})
        )
        asset = getattr(self.dataset, "asset", None)
        return normalize_sensor_tags(build_dataset_metadata, tag_list, asset=asset)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "name": self.name,
            "dataset": self.dataset.to_dict(),
            "model": self.model,
            "metadata": self.metadata.to_dict(),
            "runtime": self.runtime,
            "project_name": self.project_name,
            "evaluation": self.evaluation,
        }

    def to_json(self) -> str:
        return json.dumps(self.to_dict(), cls=MachineJSONEncoder)

    def to_yaml(self) -> str:
        return yaml.dump(
            json.loads(self.to_json()),
            Dumper=MachineSafeDumper,
            default_flow_style=False,
        )

    def report(self):
        """Run any reporters defined in runtime.reporters
        (spec: gordo/machine/machine.py:249-269)."""
        from ..reporters.base import BaseReporter
        from ..serializer import from_definition

        for reporter_def in self.runtime.get("reporters", []):
            reporter = (
                reporter_def
                if isinstance(reporter_def, BaseReporter)
                else from_definition(reporter_def)
            )
            reporter.report(self)

    def __eq__(self, other):
        return isinstance(other, Machine) and self.to_dict() == other.to_dict()

    def __repr__(self):
        return f"Machine(name={self.name!r}, project_name={self.project_name!r})"
