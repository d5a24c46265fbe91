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


def _recorded_project_name(models_dir: str) -> Optional[str]:
    """Project name the fleet builder recorded alongside the artifacts."""
    import json

    path = os.path.join(models_dir, "project.json")
    try:
        with open(path) as f:
            return json.load(f).get("project-name")
    except (OSError, ValueError):
        return None


def build_fleet_app(models_dir: str, project_name: Optional[str] = None,
                    machines: Optional[List[str]] = None):
    from fastapi import FastAPI

    if project_name is None:
        project_name = _recorded_project_name(models_dir) or "project"
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

    # Prometheus scrape endpoint: request counts/latency per machine path and
    # a resident-model gauge. Registry is per-app so repeated build_fleet_app
    # calls (tests, reloads) don't collide in the process-global registry.
    try:
        from prometheus_client import (
            CollectorRegistry,
            Counter,
            Gauge,
            Histogram,
            generate_latest,
            CONTENT_TYPE_LATEST,
        )

        registry = CollectorRegistry()
        req_count = Counter(
            "gordo_requests_total", "HTTP requests served",
            ["machine", "status"], registry=registry,
        )
        req_latency = Histogram(
            "gordo_request_seconds", "HTTP request latency",
            ["machine"], registry=registry,
            buckets=(0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 1.0, 5.0),
        )
        Gauge(
            "gordo_resident_models", "models mounted in this fleet server",
            registry=registry,
        ).set(len(mounted))

        @app.middleware("http")
        async def _observe(request, call_next):
            import time

            parts = request.url.path.split("/")
            # /gordo/v0/<project>/<machine>/... -> label by machine
            machine = parts[4] if len(parts) > 4 and parts[1] == "gordo" else "_fleet"
            t0 = time.perf_counter()
            response = await call_next(request)
            req_latency.labels(machine).observe(time.perf_counter() - t0)
            req_count.labels(machine, str(response.status_code)).inc()
            return response

        @app.get("/metrics")
        async def metrics():
            from fastapi import Response

            return Response(generate_latest(registry), media_type=CONTENT_TYPE_LATEST)
    except ImportError:  # pragma: no cover - prometheus_client not installed
        logger.info("prometheus_client unavailable; /metrics disabled")

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


def run_fleet_server(models_dir: str, project_name: Optional[str] = None,
                     host: str = "0.0.0.0", port: int = 5557):
    import uvicorn

    app = build_fleet_app(models_dir, project_name)
    uvicorn.run(app, host=host, port=port, log_level="info")
