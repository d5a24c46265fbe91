"""Multi-model fleet server.

The reference deploys one server pod per machine behind per-machine ingress
paths (/gordo/v0/<project>/<machine>/...); on one node this is a single
process serving every built model under the same path scheme, each machine's
sub-app backed by its own ModelRuntime (fused HIP forward on GPU).
"""
from __future__ import annotations

import logging
import os
from typing import Dict, List, Optional

from .. import __version__
from .app import build_app

logger = logging.getLogger(__name__)


def build_fleet_app(models_dir: str, project_name: str = "project",
                    machines: Optional[List[str]] = None):
    from fastapi import FastAPI

    if machines is None:
        machines = sorted(
            d
            for d in os.listdir(models_dir)
            if os.path.exists(os.path.join(models_dir, d, "model.pkl"))
        )
    app = FastAPI(title=f"gordo-amd fleet server [{project_name}]", version=__version__)
    mounted: Dict[str, str] = {}
    # shard resident models round-robin across the node's GPUs
    import torch

    n_gpus = torch.cuda.device_count() if torch.cuda.is_available() else 0
    for i, name in enumerate(machines):
        model_dir = os.path.join(models_dir, name)
        device = f"cuda:{i % n_gpus}" if n_gpus else None
        try:
            sub = build_app(model_dir, device=device)
        except Exception:
            logger.exception("failed to load model %s", name)
            continue
        prefix = f"/gordo/v0/{project_name}/{name}"
        app.mount(prefix, sub)
        mounted[name] = prefix

    @app.get("/healthcheck")
    async def healthcheck():
        return {"gordo-amd-version": __version__, "models": len(mounted)}

    @app.get("/")
    async def root():
        return {
            "project-name": project_name,
            "endpoints": [
                {"target-name": name, "endpoint": prefix, "healthy": True}
                for name, prefix in mounted.items()
            ],
        }

    return app


def run_fleet_server(models_dir: str, project_name: str = "project",
                     host: str = "0.0.0.0", port: int = 5557):
    import uvicorn

    app = build_fleet_app(models_dir, project_name)
    uvicorn.run(app, host=host, port=port, log_level="info")
