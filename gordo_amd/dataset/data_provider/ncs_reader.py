"""NCS-shaped archive reader.

Reader-semantics parity with gordo_components/dataset/data_provider/
ncs_reader.py (NcsReader): the reference reads ONE file per tag per YEAR from
Azure Data Lake Gen1, laid out as

    <base>/<asset dir>/<TAG.NAME>/<TAG.NAME>_<year>.(parquet|csv)

with asset routing by the tag's asset (itself inferred from the tag-name
prefix table), a thread pool fanning the per-(tag, year) reads out, missing
year files tolerated (logged + skipped — archives are ragged), and CSV files
in the semicolon `Sensor;Value;Timestamp` headerless layout. This is the
local-filesystem MI355X-native equivalent: the Azure SDK does not exist in
this offline image, but the tree layout, per-year partitioning, threading and
tolerance semantics are the reference's.
"""
from __future__ import annotations

import logging
import os
from concurrent.futures import ThreadPoolExecutor
from datetime import datetime
from typing import Iterable, List, Optional

import pandas as pd

from .base import GordoBaseDataProvider
from ...machine.sensor_tag import SensorTag, _asset_from_name

logger = logging.getLogger(__name__)


def _utc(ts) -> pd.Timestamp:
    ts = pd.Timestamp(ts)
    return ts.tz_localize("UTC") if ts.tzinfo is None else ts.tz_convert("UTC")


class NcsFileProvider(GordoBaseDataProvider):
    """Per-tag per-year sensor archives on a local filesystem tree.

    Parameters
    ----------
    base_dir:
        Root of the archive tree (the reference's data-lake store root).
    threads:
        Thread-pool width for the per-(tag, year) file reads (the reference
        threadpools per tag; file IO releases the GIL in pyarrow/pandas).
    lookup_types:
        File types tried in order per year — parquet first (the reference
        prefers parquet when present, falling back to CSV).
    """

    def __init__(
        self,
        base_dir: str,
        threads: int = 10,
        lookup_types: Optional[List[str]] = None,
    ):
        self.base_dir = base_dir
        self.threads = int(threads)
        self.lookup_types = list(lookup_types or ["parquet", "csv"])
        unknown = set(self.lookup_types) - {"parquet", "csv"}
        if unknown:
            raise ValueError(f"Unsupported lookup_types: {sorted(unknown)}")

    # ------------------------------------------------------------------
    def _asset_dir(self, tag: SensorTag) -> Optional[str]:
        asset = tag.asset or _asset_from_name(tag.name)
        if asset is None:
            return None
        return os.path.join(self.base_dir, asset)

    def can_handle_tag(self, tag: SensorTag) -> bool:
        adir = self._asset_dir(tag)
        return adir is not None and os.path.isdir(
            os.path.join(adir, tag.name.replace("/", "_"))
        )

    def _year_path(self, tag: SensorTag, year: int) -> Optional[str]:
        adir = self._asset_dir(tag)
        if adir is None:
            return None
        safe = tag.name.replace("/", "_")
        for ext in self.lookup_types:
            p = os.path.join(adir, safe, f"{safe}_{year}.{ext}")
            if os.path.exists(p):
                return p
        return None

    # ------------------------------------------------------------------
    def _read_year(self, tag: SensorTag, year: int) -> Optional[pd.Series]:
        path = self._year_path(tag, year)
        if path is None:
            # ragged archives are normal: a tag may not exist every year
            logger.debug("no %d file for tag %s", year, tag.name)
            return None
        if path.endswith(".parquet"):
            df = pd.read_parquet(path)
            ts = pd.to_datetime(df["Timestamp"], utc=True)
            vals = pd.to_numeric(df["Value"], errors="coerce")
        else:
            # reference NCS CSV layout: headerless `Sensor;Value;Timestamp`
            df = pd.read_csv(
                path,
                sep=";",
                header=None,
                names=["Sensor", "Value", "Timestamp"],
                usecols=["Value", "Timestamp"],
            )
            ts = pd.to_datetime(df["Timestamp"], utc=True, format="mixed")
            vals = pd.to_numeric(df["Value"], errors="coerce")
        s = pd.Series(vals.values, index=pd.DatetimeIndex(ts), name=tag.name)
        return s.sort_index()

    def load_series(
        self,
        from_ts: datetime,
        to_ts: datetime,
        tag_list: List[SensorTag],
        dry_run: bool = False,
    ) -> Iterable[pd.Series]:
        if from_ts >= to_ts:
            raise ValueError(f"from_ts {from_ts} must be before to_ts {to_ts}")
        from_ts, to_ts = _utc(from_ts), _utc(to_ts)
        years = range(from_ts.year, to_ts.year + 1)

        missing = [t.name for t in tag_list if self._asset_dir(t) is None]
        if missing:
            raise ValueError(
                f"Cannot route tags to an asset dir (no asset and no known "
                f"prefix): {missing[:5]}"
            )

        jobs = [(tag, year) for tag in tag_list for year in years]
        with ThreadPoolExecutor(max_workers=max(1, self.threads)) as pool:
            results = list(pool.map(lambda j: self._read_year(*j), jobs))

        n_years = len(list(years))
        for i, tag in enumerate(tag_list):
            frames = [s for s in results[i * n_years : (i + 1) * n_years] if s is not None]
            if not frames:
                raise FileNotFoundError(
                    f"No archive files for tag {tag.name!r} in years "
                    f"{list(years)} under {self._asset_dir(tag)}"
                )
            s = pd.concat(frames)
            s = s[(s.index >= from_ts) & (s.index < to_ts)]
            yield s.iloc[:1] if dry_run else s
