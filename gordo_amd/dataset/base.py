"""Dataset contract.

Parity: gordo_components/dataset/base.py — ``GordoBaseDataset`` with
``get_data() -> (X, y)`` and dict round-trip so datasets are YAML-declarable.
"""
from __future__ import annotations

import importlib
from typing import Any, Dict, Tuple, Union

import pandas as pd


class InsufficientDataError(ValueError):
    """Raised when the assembled dataset has fewer rows than required."""


class GordoBaseDataset:
    def get_data(self) -> Tuple[pd.DataFrame, Union[pd.DataFrame, None]]:
        raise NotImplementedError()

    def get_metadata(self) -> Dict[str, Any]:
        return {}

    def to_dict(self) -> Dict[str, Any]:
        cls = type(self)
        out: Dict[str, Any] = {"type": f"{cls.__module__}.{cls.__name__}"}
        for k, v in vars(self).items():
            if k.startswith("_"):
                continue
            out[k] = v
        return out

    @classmethod
    def from_dict(cls, config: Dict[str, Any]) -> "GordoBaseDataset":
        config = dict(config)
        kind = config.pop("type", "TimeSeriesDataset")
        if "." in kind:
            module_name, cls_name = kind.rsplit(".", 1)
            dataset_cls = getattr(importlib.import_module(module_name), cls_name)
        else:
            from . import datasets

            dataset_cls = getattr(datasets, kind)
        return dataset_cls(**config)


def dataset_from_config(config: Dict[str, Any]) -> GordoBaseDataset:
    return GordoBaseDataset.from_dict(config)
