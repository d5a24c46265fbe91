"""Async prediction client.

Parity: gordo_components/client/client.py — discovers a project's model
endpoints (watchman) or takes a direct endpoint, splits [start, end) into
resolution-aligned chunks (lookback-padded for sequence models), issues
bounded-concurrency async POSTs to /anomaly/prediction (parquet preferred),
retries failed chunks, reassembles DataFrames, and optionally forwards results
(client/forwarders.py).
"""
from __future__ import annotations

import asyncio
import logging
from datetime import datetime
from typing import List, Optional, Union

import aiohttp
import pandas as pd

from . import io as client_io
from .utils import EndpointMetadata, PredictionResult, make_date_ranges
from ..dataset import dataset_from_config

logger = logging.getLogger(__name__)


class Client:
    def __init__(
        self,
        project: Optional[str] = None,
        target: Optional[str] = None,
        watchman_url: Optional[str] = None,
        endpoints: Optional[List[str]] = None,
        prediction_path: str = "/anomaly/prediction",
        batch_size: int = 100000,
        parallelism: int = 10,
        forward_resampled_sensors: bool = False,
        n_retries: int = 5,
        use_parquet: bool = True,
        wire_format: str = None,
        data_provider=None,
        prediction_forwarder=None,
        max_request_hours: float = 24 * 7,
        session: Optional[aiohttp.ClientSession] = None,
    ):
        self.project = project
        self.target = target
        self.watchman_url = watchman_url
        self.direct_endpoints = endpoints or []
        self.prediction_path = prediction_path
        self.batch_size = batch_size
        self.parallelism = parallelism
        self.n_retries = n_retries
        self.use_parquet = use_parquet
        # "arrow" | "parquet" | "json"; default follows use_parquet (parity
        # with the reference client's parquet preference); "arrow" is the
        # lowest-latency option against this package's server
        self.wire_format = wire_format or ("parquet" if use_parquet else "json")
        self.data_provider = data_provider
        self.prediction_forwarder = prediction_forwarder
        self.forward_resampled_sensors = forward_resampled_sensors
        self.max_request_td = pd.Timedelta(hours=max_request_hours)
        self._session = session

    # -- discovery ---------------------------------------------------------
    async def _get_endpoints(self, session) -> List[EndpointMetadata]:
        eps: List[EndpointMetadata] = []
        if self.direct_endpoints:
            for ep in self.direct_endpoints:
                meta = await client_io.fetch_json(ep.rstrip("/") + "/metadata", session)
                name = meta.get("metadata", {}).get("name", ep)
                eps.append(EndpointMetadata(name, ep.rstrip("/"), True, meta))
        elif self.watchman_url:
            status = await client_io.fetch_json(self.watchman_url, session)
            for e in status.get("endpoints", []):
                ep_url = e.get("endpoint", "").rstrip("/")
                # the fleet server's root reports RELATIVE mount prefixes
                # (/gordo/v0/<project>/<machine>); join them onto the
                # discovery base like the reference client does with
                # watchman-reported paths
                if ep_url.startswith("/"):
                    from urllib.parse import urljoin

                    ep_url = urljoin(self.watchman_url, ep_url).rstrip("/")
                eps.append(
                    EndpointMetadata(
                        target_name=e.get("target-name", e.get("name", "?")),
                        endpoint=ep_url,
                        healthy=bool(e.get("healthy")),
                        metadata=e.get("metadata", {}),
                    )
                )
        else:
            raise ValueError("Client needs watchman_url or endpoints")
        if self.target:
            eps = [e for e in eps if e.target_name == self.target]
            if not eps:
                raise ValueError(f"Target {self.target!r} not found")
        return eps

    # -- public API ---------------------------------------------------------
    def predict(
        self,
        start: Union[str, datetime],
        end: Union[str, datetime],
        X: Optional[pd.DataFrame] = None,
    ) -> List[PredictionResult]:
        return asyncio.run(self.predict_async(start, end, X=X))

    async def predict_async(
        self,
        start: Union[str, datetime],
        end: Union[str, datetime],
        X: Optional[pd.DataFrame] = None,
    ) -> List[PredictionResult]:
        start, end = pd.Timestamp(start), pd.Timestamp(end)
        if start.tzinfo is None:
            start = start.tz_localize("UTC")
        if end.tzinfo is None:
            end = end.tz_localize("UTC")
        own_session = self._session is None
        session = self._session or aiohttp.ClientSession()
        try:
            endpoints = await self._get_endpoints(session)
            sem = asyncio.Semaphore(self.parallelism)
            tasks = [
                self._predict_single(session, sem, ep, start, end, X)
                for ep in endpoints
                if ep.healthy
            ]
            results = await asyncio.gather(*tasks)
        finally:
            if own_session:
                await session.close()
        if self.prediction_forwarder is not None:
            for res in results:
                if res.predictions is not None:
                    self.prediction_forwarder(res)
        return list(results)

    # -- internals -----------------------------------------------------------
    def _build_X(self, ep: EndpointMetadata, start, end) -> pd.DataFrame:
        """Assemble input data for [start-lookback_pad, end) via the dataset
        layer (the client pulls raw sensor data itself, like the reference)."""
        if self.data_provider is None:
            raise ValueError(
                "No data_provider given and no X passed; cannot assemble input"
            )
        resolution = ep.resolution or "10min"
        pad = pd.Timedelta(resolution) * ep.model_offset
        ds = dataset_from_config(
            {
                "type": "TimeSeriesDataset",
                "train_start_date": (start - pad).isoformat(),
                "train_end_date": end.isoformat(),
                "tag_list": ep.tag_list,
                "resolution": resolution,
                "data_provider": self.data_provider.to_dict()
                if hasattr(self.data_provider, "to_dict")
                else self.data_provider,
            }
        )
        X, _ = ds.get_data()
        return X

    async def _predict_single(
        self,
        session,
        sem: asyncio.Semaphore,
        ep: EndpointMetadata,
        start: pd.Timestamp,
        end: pd.Timestamp,
        X: Optional[pd.DataFrame],
    ) -> PredictionResult:
        from ..server.utils import (
            dataframe_from_arrow_bytes,
            dataframe_from_dict,
            dataframe_from_parquet_bytes,
            dataframe_into_arrow_bytes,
            dataframe_into_parquet_bytes,
            dataframe_to_dict,
        )

        errors: List[str] = []
        frames: List[pd.DataFrame] = []
        ranges = make_date_ranges(start, end, self.max_request_td)

        async def one_chunk(c_start, c_end):
            if X is not None:
                chunk_X = X[(X.index >= c_start) & (X.index < c_end)]
            else:
                chunk_X = await asyncio.get_event_loop().run_in_executor(
                    None, self._build_X, ep, c_start, c_end
                )
            if chunk_X.empty:
                return None
            url = ep.endpoint + self.prediction_path
            for attempt in range(self.n_retries + 1):
                try:
                    async with sem:
                        if self.wire_format == "arrow":
                            raw = await client_io.post_arrow(
                                url, session, dataframe_into_arrow_bytes(chunk_X)
                            )
                            return dataframe_from_arrow_bytes(raw)
                        if self.wire_format == "parquet":
                            raw = await client_io.post_parquet(
                                url, session, dataframe_into_parquet_bytes(chunk_X)
                            )
                            return dataframe_from_parquet_bytes(raw)
                        payload = {"X": dataframe_to_dict(chunk_X)}
                        out = await client_io.post_json(url, session, json=payload)
                        return dataframe_from_dict(out)
                except client_io.HttpUnprocessableEntity:
                    # model type mismatch — fall back to /prediction once
                    if self.prediction_path != "/prediction":
                        url = ep.endpoint + "/prediction"
                        continue
                    raise
                except (aiohttp.ClientError, IOError) as e:
                    if attempt >= self.n_retries:
                        raise
                    await asyncio.sleep(min(2**attempt * 0.1, 5))
            return None

        for c_start, c_end in ranges:
            try:
                frame = await one_chunk(c_start, c_end)
                if frame is not None:
                    frames.append(frame)
            except Exception as e:
                msg = f"{ep.target_name} [{c_start}..{c_end}): {e}"
                logger.error(msg)
                errors.append(msg)

        preds = pd.concat(frames) if frames else None
        return PredictionResult(ep.target_name, preds, errors)

    # -- misc ----------------------------------------------------------------
    def get_metadata(self) -> dict:
        async def _run():
            own = self._session is None
            session = self._session or aiohttp.ClientSession()
            try:
                eps = await self._get_endpoints(session)
                return {e.target_name: e.metadata for e in eps}
            finally:
                if own:
                    await session.close()

        return asyncio.run(_run())

    def download_model(self) -> dict:
        """target name → deserialized model object."""
        from ..serializer import loads

        async def _run():
            own = self._session is None
            session = self._session or aiohttp.ClientSession()
            out = {}
            try:
                eps = await self._get_endpoints(session)
                for e in eps:
                    raw = await client_io.fetch_bytes(e.endpoint + "/download-model", session)
                    out[e.target_name] = loads(raw)
                return out
            finally:
                if own:
                    await session.close()

        return asyncio.run(_run())
