"""Top-level helpers (spec: gordo/utils.py:15-79)."""
from __future__ import annotations

from typing import Any, Dict, List, Optional

from .core.sensor_tag import SensorTag, normalize_sensor_tag


def normalize_sensor_tags(
    build_dataset_metadata: Dict[str, Any],
    tag_list: List,
    asset: Optional[str] = None,
    **_extra_fields,
) -> List[SensorTag]:
    """
    Resolve full SensorTags for ``tag_list`` using tag metadata recorded
    in the dataset build metadata (spec: gordo/utils.py:15-50).
    Extra default-tag fields beyond ``asset`` (the reference's
    gordo-core tags carry more) are accepted and ignored — SensorTag
    here keeps only name/asset.
    """
    tags_meta: Dict[str, Dict[str, Any]] = (
        (build_dataset_metadata or {})
        .get("dataset_meta", {})
        .get("tag_loading_metadata", {})
        .get("tags", {})
    )
    # also accept the flatter layout our TimeSeriesDataset records
    if not tags_meta:
        tags_meta = (
            (build_dataset_metadata or {})
            .get("tag_loading_metadata", {})
            .get("tags", {})
        )
    out: List[SensorTag] = []
    for tag in tag_list:
        name = tag if isinstance(tag, str) else normalize_sensor_tag(tag).name
        if name in tags_meta:
            out.append(normalize_sensor_tag(tags_meta[name], asset))
        else:
            out.append(normalize_sensor_tag(tag, asset))
    return out


def join_json_paths(*paths: str) -> str:
    """Join JSON-path fragments with dots, skipping empties.

    >>> join_json_paths("a", "", "b.c")
    'a.b.c'
    """
    return ".".join(p for p in paths if p)
