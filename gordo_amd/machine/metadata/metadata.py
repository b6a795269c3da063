"""
Build-metadata dataclasses stored into metadata.json
(schema spec: gordo/machine/metadata/metadata.py:16-55).
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Any, Dict, Optional


def _asdict(obj) -> Dict[str, Any]:
    return dataclasses.asdict(obj)


@dataclass
class CrossValidationMetaData:
    scores: Dict[str, Any] = field(default_factory=dict)
    cv_duration_sec: Optional[float] = None
    splits: Dict[str, Any] = field(default_factory=dict)

    to_dict = _asdict

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "CrossValidationMetaData":
        return cls(**{k: v for k, v in (d or {}).items() if k in _field_names(cls)})


@dataclass
class ModelBuildMetadata:
    model_offset: int = 0
    model_creation_date: Optional[str] = None
    model_builder_version: Optional[str] = None
    cross_validation: CrossValidationMetaData = field(
        default_factory=CrossValidationMetaData
    )
    model_training_duration_sec: Optional[float] = None
    model_meta: Dict[str, Any] = field(default_factory=dict)

    to_dict = _asdict

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ModelBuildMetadata":
        d = dict(d or {})
        if "cross_validation" in d and isinstance(d["cross_validation"], dict):
            d["cross_validation"] = CrossValidationMetaData.from_dict(
                d["cross_validation"]
            )
        return cls(**{k: v for k, v in d.items() if k in _field_names(cls)})


@dataclass
class DatasetBuildMetadata:
    query_duration_sec: Optional[float] = None
    dataset_meta: Dict[str, Any] = field(default_factory=dict)

    to_dict = _asdict

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "DatasetBuildMetadata":
        return cls(**{k: v for k, v in (d or {}).items() if k in _field_names(cls)})


@dataclass
class BuildMetadata:
    model: ModelBuildMetadata = field(default_factory=ModelBuildMetadata)
    dataset: DatasetBuildMetadata = field(default_factory=DatasetBuildMetadata)

    to_dict = _asdict

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "BuildMetadata":
        d = dict(d or {})
        return cls(
            model=ModelBuildMetadata.from_dict(d.get("model", {})),
            dataset=DatasetBuildMetadata.from_dict(d.get("dataset", {})),
        )


@dataclass
class Metadata:
    user_defined: Dict[str, Any] = field(default_factory=dict)
    build_metadata: BuildMetadata = field(default_factory=BuildMetadata)

    to_dict = _asdict

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Metadata":
        d = dict(d or {})
        return cls(
            user_defined=d.get("user_defined", {}),
            build_metadata=BuildMetadata.from_dict(d.get("build_metadata", {})),
        )


def _field_names(cls) -> set:
    return {f.name for f in dataclasses.fields(cls)}
