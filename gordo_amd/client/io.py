"""Low-level async HTTP helpers for the prediction client.

Parity: gordo_components/client/io.py — thin aiohttp wrappers with typed
errors so the client can retry failed chunks.
"""
from __future__ import annotations

from typing import Any, Dict, Optional

import aiohttp


class HttpUnprocessableEntity(Exception):
    """422 from the server (bad payload / wrong model type)."""


class ResourceGone(Exception):
    """410 — endpoint no longer exists."""


class NotFound(Exception):
    """404."""


class BadRequest(Exception):
    """400."""


def _raise_for_status(status: int, text: str, url: str):
    if status == 422:
        raise HttpUnprocessableEntity(f"{url}: {text}")
    if status == 410:
        raise ResourceGone(f"{url}: {text}")
    if status == 404:
        raise NotFound(f"{url}: {text}")
    if status == 400:
        raise BadRequest(f"{url}: {text}")
    if status >= 400:
        raise IOError(f"{url}: HTTP {status}: {text}")


async def fetch_json(
    url: str, session: aiohttp.ClientSession, method: str = "GET", **kwargs
) -> Dict[str, Any]:
    async with session.request(method, url, **kwargs) as resp:
        if resp.status >= 400:
            _raise_for_status(resp.status, await resp.text(), url)
        return await resp.json()


async def post_json(url: str, session: aiohttp.ClientSession, json: Dict) -> Dict[str, Any]:
    return await fetch_json(url, session, method="POST", json=json)


async def post_parquet(
    url: str, session: aiohttp.ClientSession, data: bytes, params: Optional[Dict] = None
) -> bytes:
    headers = {"Content-Type": "application/octet-stream"}
    params = dict(params or {}, format="parquet")
    async with session.post(url, data=data, headers=headers, params=params) as resp:
        if resp.status >= 400:
            _raise_for_status(resp.status, await resp.text(), url)
        return await resp.read()


async def post_arrow(
    url: str, session: aiohttp.ClientSession, data: bytes, params: Optional[Dict] = None
) -> bytes:
    """Arrow IPC stream both ways — the lowest-latency wire format."""
    headers = {"Content-Type": "application/vnd.apache.arrow.stream"}
    params = dict(params or {}, format="arrow")
    async with session.post(url, data=data, headers=headers, params=params) as resp:
        if resp.status >= 400:
            _raise_for_status(resp.status, await resp.text(), url)
        return await resp.read()


async def fetch_bytes(url: str, session: aiohttp.ClientSession) -> bytes:
    async with session.get(url) as resp:
        if resp.status >= 400:
            _raise_for_status(resp.status, await resp.text(), url)
        return await resp.read()
