"""Prediction forwarders.

Parity: gordo_components/client/forwarders.py — ForwardPredictionsIntoInflux
writes each tag's prediction/anomaly columns into InfluxDB measurements.
The influxdb client is absent in this offline image, so construction is lazy
and a clear error is raised at forward time if unavailable. A CSV forwarder is
provided as the local-disk equivalent for offline pipelines.
"""
from __future__ import annotations

import logging
import os
from typing import Optional

import pandas as pd

from .utils import PredictionResult

logger = logging.getLogger(__name__)


class ForwardPredictionsIntoInflux:
    def __init__(
        self,
        destination_influx_uri: Optional[str] = None,
        destination_influx_api_key: Optional[str] = None,
        destination_influx_recreate: bool = False,
        n_retries: int = 5,
    ):
        self.uri = destination_influx_uri
        self.api_key = destination_influx_api_key
        self.recreate = destination_influx_recreate
        self.n_retries = n_retries
        self._client = None

    def _get_client(self):  # pragma: no cover - needs influxdb
        if self._client is None:
            try:
                from influxdb import DataFrameClient  # type: ignore
            except ImportError as e:
                raise RuntimeError(
                    "ForwardPredictionsIntoInflux requires the 'influxdb' "
                    "client, which is not installed in this image"
                ) from e
            from ..dataset.data_provider.providers import _parse_influx_uri

            host, port, user, pw, db = _parse_influx_uri(self.uri)
            self._client = DataFrameClient(
                host=host, port=port, username=user, password=pw, database=db
            )
            if self.recreate:
                self._client.drop_database(db)
                self._client.create_database(db)
        return self._client

    def __call__(self, result: PredictionResult):  # pragma: no cover
        client = self._get_client()
        df = result.predictions
        if df is None or df.empty:
            return
        for top in df.columns.get_level_values(0).unique():
            sub = df[top]
            if isinstance(sub, pd.Series):
                sub = sub.to_frame(name="value")
            measurement = f"{result.name}.{top}"
            client.write_points(sub, measurement)


class ForwardPredictionsToDisk:
    """Write each machine's prediction frame as parquet under a directory —
    the offline stand-in for the Influx forwarder."""

    def __init__(self, destination_dir: str):
        self.destination_dir = destination_dir
        os.makedirs(destination_dir, exist_ok=True)

    def __call__(self, result: PredictionResult):
        if result.predictions is None or result.predictions.empty:
            return
        from ..server.utils import dataframe_into_parquet_bytes

        path = os.path.join(self.destination_dir, f"{result.name}.parquet")
        with open(path, "wb") as f:
            f.write(dataframe_into_parquet_bytes(result.predictions))
        logger.info("Forwarded %s predictions to %s", result.name, path)
