"""Machine & NormalizedConfig — the project-config layer.

Capability parity with the reference's workflow config elements
(gordo_components/workflow/config_elements/machine.py and
normalized_config.py): a project YAML holds a ``machines:`` list and a
``globals:`` mapping of defaults; each Machine merges its own config over the
globals, producing a fully-specified unit of work (name + dataset + model +
metadata + evaluation + runtime).

The MI355X-native difference is only downstream: instead of compiling to an
Argo workflow of per-machine pods, Machines feed the in-node fleet scheduler
(gordo_amd.workflow.fleet) that packs builds across the 8 GPUs of one node.
"""
from __future__ import annotations

import copy
from typing import Any, Dict, List, Optional

import yaml

from .sensor_tag import normalize_sensor_tags

# Default model definition used when a machine/global config does not name one
# (parity: NormalizedConfig.DEFAULT_CONFIG in the reference, which defaults to
# KerasAutoEncoder(kind=feedforward_hourglass) inside a MinMaxScaler pipeline
# wrapped by DiffBasedAnomalyDetector).
DEFAULT_MODEL_CONFIG: Dict[str, Any] = {
    "gordo_amd.model.anomaly.diff.DiffBasedAnomalyDetector": {
        "base_estimator": {
            "sklearn.pipeline.Pipeline": {
                "steps": [
                    "sklearn.preprocessing.MinMaxScaler",
                    {
                        "gordo_amd.model.models.TorchAutoEncoder": {
                            "kind": "feedforward_hourglass"
                        }
                    },
                ]
            }
        }
    }
}

DEFAULT_EVALUATION_CONFIG: Dict[str, Any] = {
    "cv_mode": "full_build",
    "scoring_scaler": "sklearn.preprocessing.RobustScaler",
    "metrics": ["explained_variance_score", "r2_score", "mean_squared_error", "mean_absolute_error"],
}

DEFAULT_RUNTIME_CONFIG: Dict[str, Any] = {
    # MI355X-native runtime knobs (replaces the reference's k8s resource block)
    "device": "auto",  # "auto" | "cuda:N" | "cpu"
    "dtype": "bf16",
    "batch_size": 8192,
    "max_epochs": 20,
}


def _deep_merge(base: Dict[str, Any], override: Dict[str, Any]) -> Dict[str, Any]:
    """Recursively merge override over base (dicts only; other values replace)."""
    out = copy.deepcopy(base)
    for k, v in override.items():
        if k in out and isinstance(out[k], dict) and isinstance(v, dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


class Machine:
    """One named model bound to one dataset config and one model config."""

    def __init__(
        self,
        name: str,
        dataset: Dict[str, Any],
        model: Optional[Dict[str, Any]] = None,
        metadata: Optional[Dict[str, Any]] = None,
        evaluation: Optional[Dict[str, Any]] = None,
        runtime: Optional[Dict[str, Any]] = None,
        project_name: Optional[str] = None,
    ):
        if not name or "/" in name or " " in name:
            raise ValueError(f"Invalid machine name: {name!r}")
        self.name = name
        self.dataset = dataset
        self.model = model if model is not None else copy.deepcopy(DEFAULT_MODEL_CONFIG)
        self.metadata = metadata or {}
        self.evaluation = _deep_merge(DEFAULT_EVALUATION_CONFIG, evaluation or {})
        self.runtime = _deep_merge(DEFAULT_RUNTIME_CONFIG, runtime or {})
        self.project_name = project_name

        # Normalize tags eagerly so config errors surface at compile time
        if "tag_list" in self.dataset:
            asset = self.dataset.get("asset")
            self.dataset["tag_list"] = [
                t.to_json()
                for t in normalize_sensor_tags(self.dataset["tag_list"], asset)
            ]
        if "target_tag_list" in self.dataset and self.dataset["target_tag_list"]:
            asset = self.dataset.get("asset")
            self.dataset["target_tag_list"] = [
                t.to_json()
                for t in normalize_sensor_tags(self.dataset["target_tag_list"], asset)
            ]

    @classmethod
    def from_config(
        cls,
        config: Dict[str, Any],
        project_name: Optional[str] = None,
        config_globals: Optional[Dict[str, Any]] = None,
    ) -> "Machine":
        g = config_globals or {}
        name = config.get("name")
        if name is None:
            raise ValueError(f"Machine config missing 'name': {config!r}")
        dataset = _deep_merge(g.get("dataset", {}), config.get("dataset", {}))
        model = config.get("model", g.get("model"))
        metadata = _deep_merge(
            g.get("metadata", {}), config.get("metadata", {}) or {}
        )
        evaluation = _deep_merge(g.get("evaluation", {}), config.get("evaluation", {}))
        runtime = _deep_merge(g.get("runtime", {}), config.get("runtime", {}))
        return cls(
            name=name,
            dataset=dataset,
            model=model,
            metadata=metadata,
            evaluation=evaluation,
            runtime=runtime,
            project_name=project_name,
        )

    def to_dict(self) -> Dict[str, Any]:
        return {
            "name": self.name,
            "dataset": self.dataset,
            "model": self.model,
            "metadata": self.metadata,
            "evaluation": self.evaluation,
            "runtime": self.runtime,
            "project_name": self.project_name,
        }

    def __repr__(self):
        return f"Machine(name={self.name!r})"


class NormalizedConfig:
    """Parse a project YAML into a list of fully-merged Machines.

    Parity: gordo_components/workflow/config_elements/normalized_config.py —
    ``globals:`` defaults merged under each entry of ``machines:``.
    """

    def __init__(self, config: Dict[str, Any], project_name: Optional[str] = None):
        # Explicit argument wins; otherwise a top-level ``project_name:`` key
        # in the YAML itself; otherwise the reference's "project" default.
        project_name = project_name or config.get("project_name") or "project"
        self.project_name = project_name
        self.globals: Dict[str, Any] = config.get("globals") or {}
        machines = config.get("machines")
        if not machines:
            raise ValueError("Project config has no 'machines' list")
        names = [m.get("name") for m in machines]
        dupes = {n for n in names if names.count(n) > 1}
        if dupes:
            raise ValueError(f"Duplicate machine names: {sorted(dupes)}")
        self.machines: List[Machine] = [
            Machine.from_config(m, project_name=project_name, config_globals=self.globals)
            for m in machines
        ]

    @classmethod
    def from_yaml(cls, text: str, project_name: Optional[str] = None) -> "NormalizedConfig":
        return cls(yaml.safe_load(text), project_name=project_name)
