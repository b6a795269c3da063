"""
SensorTag — the unit naming one timeseries tag.

Behavioral spec from the reference's external gordo_core package
(import sites: gordo/machine/machine.py:9-11, gordo/utils.py:15-50).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Union, Dict, Any

from .exceptions import SensorTagNormalizationError


@dataclass(frozen=True)
class SensorTag:
    name: str
    asset: Optional[str] = field(default=None)

    def to_json(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"name": self.name}
        if self.asset is not None:
            d["asset"] = self.asset
        return d

    @classmethod
    def from_json(cls, obj: Union[str, Dict[str, Any], List]) -> "SensorTag":
        return normalize_sensor_tag(obj)


Tag = Union[str, Dict[str, Any], List, SensorTag]


def normalize_sensor_tag(tag: Tag, asset: Optional[str] = None) -> SensorTag:
    """
    Coerce any accepted tag representation into a ``SensorTag``.

    Accepted forms: ``SensorTag``, ``str``, ``{"name": ..., "asset": ...}``,
    ``[name, asset]``.

    Examples
    --------
    >>> normalize_sensor_tag("Tag 1")
    SensorTag(name='Tag 1', asset=None)
    >>> normalize_sensor_tag({"name": "Tag 1", "asset": "a"})
    SensorTag(name='Tag 1', asset='a')
    """
    if isinstance(tag, SensorTag):
        if asset is not None and tag.asset is None:
            return SensorTag(tag.name, asset)
        return tag
    if isinstance(tag, str):
        return SensorTag(tag, asset)
    if isinstance(tag, dict):
        if "name" not in tag:
            raise SensorTagNormalizationError(
                f"Sensor tag dict {tag!r} has no 'name' key"
            )
        return SensorTag(str(tag["name"]), tag.get("asset", asset))
    if isinstance(tag, (list, tuple)):
        if not tag:
            raise SensorTagNormalizationError("Empty sensor tag list")
        name = str(tag[0])
        tag_asset = str(tag[1]) if len(tag) > 1 and tag[1] is not None else asset
        return SensorTag(name, tag_asset)
    raise SensorTagNormalizationError(
        f"Unable to normalize sensor tag of type {type(tag)}: {tag!r}"
    )


def extract_tag_name(tag: Tag) -> str:
    if isinstance(tag, SensorTag):
        return tag.name
    if isinstance(tag, str):
        return tag
    return normalize_sensor_tag(tag).name


def unique_tag_names(*tag_lists) -> Dict[str, SensorTag]:
    """Map tag-name -> SensorTag across several tag lists (dedup by name)."""
    out: Dict[str, SensorTag] = {}
    for tags in tag_lists:
        if tags is None:
            continue
        for t in tags:
            st = normalize_sensor_tag(t)
            out.setdefault(st.name, st)
    return out
