"""Sensor tag normalization.

Capability parity with gordo_components/dataset/sensor_tag.py: map the YAML
forms a tag may take (plain string, [name, asset] list, {name:, asset:} dict,
SensorTag) onto a single SensorTag(name, asset) value type.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Iterable, List, Optional, Union


class SensorTagNormalizationError(ValueError):
    """Raised when a raw tag specification cannot be normalized."""


@dataclass(frozen=True)
class SensorTag:
    name: str
    asset: Optional[str] = None

    def to_json(self) -> dict:
        return {"name": self.name, "asset": self.asset}


# Known asset prefixes (reference keeps a table mapping tag prefixes to
# assets, e.g. "asgb." → asset codes; we keep the mechanism, with a small
# default table that callers may extend).
ASSET_BY_PREFIX = {
    "asgb": "1191-asgb",
    "gra": "1755-gra",
    "1125": "1125-kvb",
    "trb": "1775-trd",
    "trc": "1776-trc",
    "tra": "1130-troa",
    "per": "1900-per",
}


def _asset_from_name(name: str) -> Optional[str]:
    head = name.split(".")[0].lower()
    for prefix, asset in ASSET_BY_PREFIX.items():
        if head.startswith(prefix):
            return asset
    return None


TagSpec = Union[str, dict, list, tuple, SensorTag]


def normalize_sensor_tag(tag: TagSpec, asset: Optional[str] = None) -> SensorTag:
    """Normalize one raw tag spec to a SensorTag.

    Accepted forms (parity with reference ``normalize_sensor_tags``):
      - SensorTag (passthrough)
      - "TAG.NAME"                          (asset inferred from prefix or arg)
      - ["TAG.NAME", "asset"] / tuple
      - {"name": "TAG.NAME", "asset": "asset"}
    """
    if isinstance(tag, SensorTag):
        return tag
    if isinstance(tag, str):
        return SensorTag(name=tag, asset=asset or _asset_from_name(tag))
    if isinstance(tag, dict):
        if "name" not in tag:
            raise SensorTagNormalizationError(f"Tag dict missing 'name': {tag!r}")
        return SensorTag(name=tag["name"], asset=tag.get("asset", asset))
    if isinstance(tag, (list, tuple)):
        if len(tag) == 2:
            return SensorTag(name=str(tag[0]), asset=str(tag[1]))
        if len(tag) == 1:
            return normalize_sensor_tag(tag[0], asset)
        raise SensorTagNormalizationError(f"Tag list must have 1-2 items: {tag!r}")
    raise SensorTagNormalizationError(f"Unrecognized tag spec: {tag!r}")


def normalize_sensor_tags(
    tags: Iterable[TagSpec], asset: Optional[str] = None
) -> List[SensorTag]:
    return [normalize_sensor_tag(t, asset) for t in tags]


def to_list_of_strings(tags: Iterable[SensorTag]) -> List[str]:
    return [t.name for t in tags]
