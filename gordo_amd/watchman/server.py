"""Watchman — per-project fleet supervisor.

Parity: gordo_components/watchman/ — builds the expected endpoint list from
the project config, polls each model server's /healthcheck + /metadata in the
background, caches statuses, serves the aggregate at GET / (what the client
uses for discovery).
"""
from __future__ import annotations

import asyncio
import logging
import threading
import time
from typing import Dict, List, Optional

from .. import __version__

logger = logging.getLogger(__name__)


class EndpointStatuses:
    def __init__(self, targets: Dict[str, str], refresh_interval: float = 10.0):
        """targets: machine name → base endpoint URL."""
        self.targets = targets
        self.refresh_interval = refresh_interval
        self._statuses: Dict[str, dict] = {
            name: {
                "target-name": name,
                "endpoint": url,
                "healthy": False,
                "metadata": {},
                "last-checked": None,
            }
            for name, url in targets.items()
        }
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self):
        self._thread = threading.Thread(target=self._poll_loop, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()

    def _poll_loop(self):
        while not self._stop.is_set():
            try:
                asyncio.run(self.refresh())
            except Exception:
                logger.exception("watchman refresh failed")
            self._stop.wait(self.refresh_interval)

    async def refresh(self):
        import aiohttp

        async with aiohttp.ClientSession(
            timeout=aiohttp.ClientTimeout(total=5)
        ) as session:
            await asyncio.gather(
                *(self._check(session, name, url) for name, url in self.targets.items())
            )

    async def _check(self, session, name: str, url: str):
        status = {
            "target-name": name,
            "endpoint": url,
            "healthy": False,
            "metadata": {},
            "last-checked": time.time(),
        }
        try:
            async with session.get(url.rstrip("/") + "/healthcheck") as r:
                status["healthy"] = r.status == 200
            if status["healthy"]:
                async with session.get(url.rstrip("/") + "/metadata") as r:
                    if r.status == 200:
                        status["metadata"] = await r.json()
        except Exception as e:
            status["error"] = str(e)
        with self._lock:
            self._statuses[name] = status

    def snapshot(self) -> List[dict]:
        with self._lock:
            return list(self._statuses.values())


def build_app(project_name: str, targets: Dict[str, str], refresh_interval: float = 10.0):
    from contextlib import asynccontextmanager

    from fastapi import FastAPI

    statuses = EndpointStatuses(targets, refresh_interval)

    @asynccontextmanager
    async def lifespan(app):
        # lifespan context (the on_event decorators are deprecated)
        statuses.start()
        try:
            yield
        finally:
            statuses.stop()

    app = FastAPI(
        title=f"gordo-amd watchman [{project_name}]",
        version=__version__,
        lifespan=lifespan,
    )
    app.state.statuses = statuses

    @app.get("/")
    async def root():
        eps = statuses.snapshot()
        return {
            "project-name": project_name,
            "gordo-amd-version": __version__,
            "endpoints": eps,
            "project-healthy": all(e["healthy"] for e in eps) if eps else False,
        }

    @app.get("/healthcheck")
    async def healthcheck():
        return {"gordo-amd-version": __version__}

    return app


def run_watchman(
    project_name: str,
    targets: Dict[str, str],
    host: str = "0.0.0.0",
    port: int = 5556,
    refresh_interval: float = 10.0,
):
    import uvicorn

    app = build_app(project_name, targets, refresh_interval)
    uvicorn.run(app, host=host, port=port, log_level="info")
