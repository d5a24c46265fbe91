"""YAML/dict definition → live object graph.

Parity: gordo_components/serializer/pipeline_from_definition.py — each key is
a dotted import path, its value the kwargs; sklearn ``Pipeline.steps`` /
``FeatureUnion.transformer_list`` and nested estimator definitions are
resolved recursively. This is what makes the whole model zoo YAML-extensible
without code changes.
"""
from __future__ import annotations

import importlib
import logging
from typing import Any, Union

logger = logging.getLogger(__name__)


def _import_from_path(path: str):
    module_name, _, attr = path.rpartition(".")
    if not module_name:
        raise ImportError(f"Not a dotted path: {path!r}")
    module = importlib.import_module(module_name)
    return getattr(module, attr)


def _looks_like_definition(value: Any) -> bool:
    """A dict with exactly one dotted-path key whose target imports."""
    if isinstance(value, str) and "." in value:
        try:
            _import_from_path(value)
            return True
        except Exception:
            return False
    if isinstance(value, dict) and len(value) == 1:
        key = next(iter(value))
        if isinstance(key, str) and "." in key:
            try:
                _import_from_path(key)
                return True
            except Exception:
                return False
    return False


def _resolve_value(value: Any) -> Any:
    if _looks_like_definition(value):
        return from_definition(value)
    if isinstance(value, list):
        return [_resolve_value(v) for v in value]
    if isinstance(value, dict):
        return {k: _resolve_value(v) for k, v in value.items()}
    return value


def _step_name(i: int, obj: Any) -> str:
    return f"step_{i}_{type(obj).__name__.lower()}"


def _resolve_steps(steps: list) -> list:
    """Pipeline steps / FeatureUnion transformer_list: list of definitions →
    list of (name, object) tuples."""
    out = []
    for i, step in enumerate(steps):
        if (
            isinstance(step, (list, tuple))
            and len(step) == 2
            and isinstance(step[0], str)
            and not _looks_like_definition(step[0])
        ):
            name, inner = step
            obj = from_definition(inner)
        else:
            obj = from_definition(step)
            name = _step_name(i, obj)
        out.append((name, obj))
    return out


def from_definition(definition: Union[str, dict]) -> Any:
    """Instantiate the object a definition describes."""
    if isinstance(definition, str):
        cls = _import_from_path(definition)
        return cls()
    if not isinstance(definition, dict) or len(definition) != 1:
        raise ValueError(
            f"Definition must be a dotted-path string or single-key dict, "
            f"got: {definition!r}"
        )
    path, params = next(iter(definition.items()))
    cls = _import_from_path(path)
    if params is None:
        return cls()
    if not isinstance(params, dict):
        raise ValueError(f"Params for {path} must be a mapping, got {params!r}")

    kwargs = {}
    for key, value in params.items():
        if key in ("steps", "transformer_list") and isinstance(value, list):
            kwargs[key] = _resolve_steps(value)
        elif key == "func" and isinstance(value, str) and "." in value:
            # FunctionTransformer(func="gordo_amd.model.transformer_funcs.general.multiply")
            kwargs[key] = _import_from_path(value)
        else:
            kwargs[key] = _resolve_value(value)
    return cls(**kwargs)


# Back-compat alias matching the reference's public name
pipeline_from_definition = from_definition
