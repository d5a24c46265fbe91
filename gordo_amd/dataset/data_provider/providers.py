"""Concrete data providers.

Parity targets (gordo_components/dataset/data_provider/providers.py,
ncs_reader.py, iroc_reader.py):

- RandomDataProvider — THE fake backend the whole reference test-suite runs on
  (synthetic per-tag series; no cloud, no cluster).
- InfluxDataProvider — queries one measurement per tag from InfluxDB; gated on
  the optional ``influxdb`` client being importable (no network in this image).
- ParquetFileProvider / CsvFileProvider — local-file equivalents of the
  reference's NCS (one file per tag) readers; these cover the "read sensor
  archives from disk" capability without the Azure Data Lake SDK, which does
  not exist in this offline image.
"""
from __future__ import annotations

import os
import zlib
from datetime import datetime
from typing import Iterable, List, Optional

import numpy as np
import pandas as pd

from .base import GordoBaseDataProvider
from ...machine.sensor_tag import SensorTag


class RandomDataProvider(GordoBaseDataProvider):
    """Deterministic synthetic sensor data (seeded per tag name)."""

    def __init__(self, min_size: int = 100, max_size: int = 300, resolution: str = "10min"):
        self.min_size = min_size
        self.max_size = max_size
        self.resolution = resolution

    def can_handle_tag(self, tag: SensorTag) -> bool:
        return True

    def load_series(
        self,
        from_ts: datetime,
        to_ts: datetime,
        tag_list: List[SensorTag],
        dry_run: bool = False,
    ) -> Iterable[pd.Series]:
        if from_ts >= to_ts:
            raise ValueError(f"from_ts {from_ts} must be before to_ts {to_ts}")
        index = pd.date_range(start=from_ts, end=to_ts, freq=self.resolution)
        for tag in tag_list:
            # zlib.crc32 is stable across processes; hash() is salted per
            # interpreter and would make "deterministic" data run-dependent.
            rng = np.random.default_rng(zlib.crc32(tag.name.encode()))
            n = min(len(index), int(rng.integers(self.min_size, self.max_size + 1)))
            # walkish signal so scaling / anomaly scoring has structure
            vals = np.cumsum(rng.standard_normal(n) * 0.1) + rng.normal(0, 1)
            series = pd.Series(vals, index=index[:n], name=tag.name)
            if dry_run:
                series = series.iloc[:1]
            yield series


class InfluxDataProvider(GordoBaseDataProvider):
    """Per-tag measurement reads from InfluxDB (parity: InfluxDataProvider).

    The influxdb client is not installed in this offline image; construction
    succeeds (so configs parse) but load_series raises a clear error unless
    the client is importable.
    """

    def __init__(
        self,
        measurement: str = "sensor_data",
        value_name: str = "Value",
        api_key: Optional[str] = None,
        api_key_header: Optional[str] = None,
        uri: Optional[str] = None,
        **influx_kwargs,
    ):
        self.measurement = measurement
        self.value_name = value_name
        self.api_key = api_key
        self.api_key_header = api_key_header
        self.uri = uri
        self._influx_kwargs = influx_kwargs

    def can_handle_tag(self, tag: SensorTag) -> bool:
        return True

    def _client(self):
        try:
            from influxdb import DataFrameClient  # type: ignore
        except ImportError as e:  # pragma: no cover - no influx in image
            raise RuntimeError(
                "InfluxDataProvider requires the 'influxdb' client, which is "
                "not installed in this image"
            ) from e
        kwargs = dict(self._influx_kwargs)
        if self.uri:
            host, port, username, password, database = _parse_influx_uri(self.uri)
            kwargs.update(
                host=host, port=port, username=username, password=password, database=database
            )
        if self.api_key:
            kwargs.setdefault("headers", {})[self.api_key_header or "Ocp-Apim-Subscription-Key"] = self.api_key
        return DataFrameClient(**kwargs)

    def load_series(self, from_ts, to_ts, tag_list, dry_run=False):  # pragma: no cover
        client = self._client()
        for tag in tag_list:
            q = (
                f'SELECT "{self.value_name}" FROM "{self.measurement}" WHERE '
                f"\"tag\" = '{tag.name}' AND time >= '{from_ts.isoformat()}' "
                f"AND time < '{to_ts.isoformat()}'"
            )
            res = client.query(q)
            df = res.get(self.measurement, pd.DataFrame(columns=[self.value_name]))
            s = df[self.value_name]
            s.name = tag.name
            yield s


def _parse_influx_uri(uri: str):
    """'<user>:<pw>@<host>:<port>/<db>' -> parts (parity: influx uri parsing)."""
    creds, _, rest = uri.rpartition("@")
    username, _, password = creds.partition(":")
    hostport, _, database = rest.partition("/")
    host, _, port = hostport.partition(":")
    return host, int(port or 8086), username or None, password or None, database or None


class _FileProvider(GordoBaseDataProvider):
    """Base for one-file-per-tag local archives (MI355X-native stand-in for the
    reference's NcsReader which reads one parquet/CSV per tag per year from
    Azure Data Lake Gen1)."""

    extension = ""

    def __init__(self, base_dir: str, timestamp_column: str = "timestamp", value_column: str = "value"):
        self.base_dir = base_dir
        self.timestamp_column = timestamp_column
        self.value_column = value_column

    def _path_for(self, tag: SensorTag) -> str:
        safe = tag.name.replace("/", "_")
        return os.path.join(self.base_dir, f"{safe}{self.extension}")

    def can_handle_tag(self, tag: SensorTag) -> bool:
        return os.path.exists(self._path_for(tag))

    def _read(self, path: str) -> pd.DataFrame:
        raise NotImplementedError()

    def load_series(self, from_ts, to_ts, tag_list, dry_run=False):
        for tag in tag_list:
            path = self._path_for(tag)
            if not os.path.exists(path):
                raise FileNotFoundError(f"No archive for tag {tag.name!r} at {path}")
            df = self._read(path)
            ts = pd.to_datetime(df[self.timestamp_column], utc=True)
            s = pd.Series(df[self.value_column].values, index=ts, name=tag.name)
            s = s[(s.index >= pd.Timestamp(from_ts)) & (s.index < pd.Timestamp(to_ts))]
            yield s.iloc[:1] if dry_run else s


class ParquetFileProvider(_FileProvider):
    extension = ".parquet"

    def _read(self, path: str) -> pd.DataFrame:
        return pd.read_parquet(path)


class CsvFileProvider(_FileProvider):
    extension = ".csv"

    def _read(self, path: str) -> pd.DataFrame:
        return pd.read_csv(path)
