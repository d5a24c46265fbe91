"""Data provider contract.

Parity: gordo_components/dataset/data_provider/base.py —
``load_series(from_ts, to_ts, tag_list) -> Iterable[pd.Series]`` plus
dict round-trip (``from_dict`` / ``to_dict``) so providers are YAML-declarable.
"""
from __future__ import annotations

import importlib
from datetime import datetime
from typing import Iterable, List

import pandas as pd

from ...machine.sensor_tag import SensorTag


class GordoBaseDataProvider:
    def load_series(
        self,
        from_ts: datetime,
        to_ts: datetime,
        tag_list: List[SensorTag],
        dry_run: bool = False,
    ) -> Iterable[pd.Series]:
        raise NotImplementedError()

    def can_handle_tag(self, tag: SensorTag) -> bool:
        raise NotImplementedError()

    def to_dict(self) -> dict:
        cls = type(self)
        return {
            "type": f"{cls.__module__}.{cls.__name__}",
            **{k: v for k, v in vars(self).items() if not k.startswith("_")},
        }

    @classmethod
    def from_dict(cls, config: dict) -> "GordoBaseDataProvider":
        config = dict(config)
        kind = config.pop("type", None)
        if kind is None:
            # default provider
            from .providers import RandomDataProvider

            return RandomDataProvider(**config)
        if "." in kind:
            module_name, cls_name = kind.rsplit(".", 1)
            module = importlib.import_module(module_name)
            provider_cls = getattr(module, cls_name)
        else:
            # short names resolve against the package namespace (providers,
            # ncs_reader, iroc_reader re-exports)
            pkg = importlib.import_module(__package__)
            provider_cls = getattr(pkg, kind)
        return provider_cls(**config)
