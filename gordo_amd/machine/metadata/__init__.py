from .metadata import (
    Metadata,
    BuildMetadata,
    ModelBuildMetadata,
    DatasetBuildMetadata,
    CrossValidationMetaData,
)

__all__ = [
    "Metadata",
    "BuildMetadata",
    "ModelBuildMetadata",
    "DatasetBuildMetadata",
    "CrossValidationMetaData",
]
