"""Workflow generator.

Parity: gordo_components/workflow/workflow_generator/ — the reference renders
project YAML into an Argo Workflow (one builder pod per machine + server
Deployments + watchman). The MI355X-native equivalent compiles the same
project YAML into a node-local deployment plan: the fleet builder's task list
plus per-machine server specs (ports/paths) and the watchman target map —
still emitted as YAML so operators can inspect/check it in, and consumable by
`gordo-amd fleet` commands.
"""
from __future__ import annotations

from typing import Any, Dict, Optional

import yaml

from .. import __version__
from ..machine import NormalizedConfig


def generate_plan(
    config: NormalizedConfig,
    project_name: str = "project",
    output_dir: str = "./models",
    registry_dir: Optional[str] = None,
    base_port: int = 30000,
    n_gpus: int = 8,
) -> Dict[str, Any]:
    machines = config.machines
    builder_tasks = []
    servers = []
    watchman_targets = {}
    for i, m in enumerate(machines):
        device = f"cuda:{i % n_gpus}" if n_gpus else "cpu"
        builder_tasks.append(
            {
                "name": m.name,
                "device": device,
                "output-dir": f"{output_dir}/{m.name}",
                "registry-dir": registry_dir,
                "machine-config": m.to_dict(),
            }
        )
        port = base_port + i
        url = f"http://127.0.0.1:{port}"
        servers.append(
            {
                "name": m.name,
                "model-dir": f"{output_dir}/{m.name}",
                "port": port,
                "paths": ["/prediction", "/anomaly/prediction", "/metadata"],
            }
        )
        watchman_targets[m.name] = url
    return {
        "apiVersion": "gordo-amd/v1",
        "kind": "NodePlan",
        "metadata": {"project-name": project_name, "generator-version": __version__},
        "spec": {
            "builder": {"tasks": builder_tasks, "n-gpus": n_gpus},
            "servers": servers,
            "watchman": {"targets": watchman_targets, "port": base_port - 1},
        },
    }


def generate_yaml(config: NormalizedConfig, **kwargs) -> str:
    return yaml.safe_dump(generate_plan(config, **kwargs), default_flow_style=False)


def unique_tags(config: NormalizedConfig) -> list:
    """All distinct sensor tags across the project (parity: the reference's
    `workflow unique-tags` command)."""
    seen = []
    for m in config.machines:
        for t in m.dataset.get("tag_list", []):
            name = t["name"] if isinstance(t, dict) else str(t)
            if name not in seen:
                seen.append(name)
    return seen
