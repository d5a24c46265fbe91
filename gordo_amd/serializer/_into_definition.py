"""Live object graph → YAML-able definition.

Parity: gordo_components/serializer/pipeline_into_definition.py — the inverse
walk of from_definition via ``get_params()``, so that
``into_definition(from_definition(d))`` round-trips.
"""
from __future__ import annotations

import logging
from typing import Any

import numpy as np
from sklearn.base import BaseEstimator
from sklearn.pipeline import FeatureUnion, Pipeline

logger = logging.getLogger(__name__)


def _dotted_path(obj: Any) -> str:
    cls = type(obj)
    return f"{cls.__module__}.{cls.__qualname__}"


def _plain(value: Any) -> Any:
    """Convert a param value into a YAML-safe representation."""
    if isinstance(value, (Pipeline, FeatureUnion)):
        return into_definition(value)
    if isinstance(value, BaseEstimator):
        return into_definition(value)
    if callable(value) and hasattr(value, "__module__") and hasattr(value, "__name__"):
        return f"{value.__module__}.{value.__name__}"
    if isinstance(value, np.generic):
        return value.item()
    if isinstance(value, np.ndarray):
        return value.tolist()
    if isinstance(value, tuple):
        return [_plain(v) for v in value]
    if isinstance(value, list):
        return [_plain(v) for v in value]
    if isinstance(value, dict):
        return {k: _plain(v) for k, v in value.items()}
    return value


def into_definition(obj: Any, prune_default_params: bool = False) -> Any:
    """Definition dict for obj (single-key: dotted path → params)."""
    if isinstance(obj, Pipeline):
        steps = [into_definition(step) for _, step in obj.steps]
        params = {"steps": steps}
        extra = {
            k: _plain(v)
            for k, v in obj.get_params(deep=False).items()
            if k not in ("steps", "memory", "verbose") and v is not None
        }
        params.update(extra)
        return {_dotted_path(obj): params}
    if isinstance(obj, FeatureUnion):
        transformers = [into_definition(t) for _, t in obj.transformer_list]
        params = {"transformer_list": transformers}
        return {_dotted_path(obj): params}
    if isinstance(obj, BaseEstimator) or hasattr(obj, "get_params"):
        params = {}
        for k, v in obj.get_params(deep=False).items():
            if prune_default_params and v is None:
                continue
            params[k] = _plain(v)
        return {_dotted_path(obj): params}
    raise TypeError(f"Cannot build a definition for {obj!r}")


pipeline_into_definition = into_definition
