"""Model-factory registry.

Parity: gordo_components/model/register.py — ``register_model_builder`` maps a
``kind`` string (what YAML names, e.g. ``feedforward_hourglass``) to a factory
callable. Factories must take ``n_features`` and ``n_features_out`` as their
first args; validated at registration time.
"""
from __future__ import annotations

import inspect
from typing import Callable, Dict

FACTORIES: Dict[str, Dict[str, Callable]] = {}


class register_model_builder:
    """Decorator: @register_model_builder(type="TorchAutoEncoder")."""

    def __init__(self, type: str):
        self.type = type

    def __call__(self, build_fn: Callable) -> Callable:
        self._validate(build_fn)
        FACTORIES.setdefault(self.type, {})[build_fn.__name__] = build_fn
        return build_fn

    @staticmethod
    def _validate(build_fn: Callable):
        params = list(inspect.signature(build_fn).parameters)
        required = ["n_features", "n_features_out"]
        if params[: len(required)] != required:
            raise ValueError(
                f"Factory {build_fn.__name__} must take {required} as its "
                f"first parameters, got {params[:2]}"
            )


def lookup_factory(model_type: str, kind: str) -> Callable:
    """Resolve a factory; accepts a dotted path as an escape hatch."""
    for type_name, kinds in FACTORIES.items():
        if model_type == type_name or model_type.endswith(type_name):
            if kind in kinds:
                return kinds[kind]
            raise ValueError(
                f"Unknown kind {kind!r} for {type_name}; registered: {sorted(kinds)}"
            )
    # Fall back: kind may itself be a dotted import path to a factory
    if "." in kind:
        import importlib

        module_name, fn_name = kind.rsplit(".", 1)
        return getattr(importlib.import_module(module_name), fn_name)
    raise ValueError(f"No factories registered for model type {model_type!r}")
