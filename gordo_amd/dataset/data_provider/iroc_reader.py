"""IROC-shaped batched-CSV reader.

Reader-semantics parity with gordo_components/dataset/data_provider/
iroc_reader.py (IrocReader): IROC archives are BATCHED CSVs — each file holds
many tags' readings as (tag, value, timestamp) rows — grouped under per-asset
directories. The reference threadpools the file reads, pivots rows out to one
series per requested tag, and warns about requested tags with no rows. Local
filesystem equivalent of the reference's data-lake paths:

    <base>/<asset dir>/*.csv     with header  tag,value,timestamp
"""
from __future__ import annotations

import glob
import logging
import os
from concurrent.futures import ThreadPoolExecutor
from datetime import datetime
from typing import Iterable, List

import pandas as pd

from .base import GordoBaseDataProvider
from .ncs_reader import _utc
from ...machine.sensor_tag import SensorTag, _asset_from_name

logger = logging.getLogger(__name__)


class IrocFileProvider(GordoBaseDataProvider):
    def __init__(self, base_dir: str, threads: int = 10):
        self.base_dir = base_dir
        self.threads = int(threads)

    def _asset_dir(self, tag: SensorTag) -> str:
        asset = tag.asset or _asset_from_name(tag.name)
        if asset is None:
            raise ValueError(f"Cannot route tag {tag.name!r} to an asset dir")
        return os.path.join(self.base_dir, asset)

    def can_handle_tag(self, tag: SensorTag) -> bool:
        try:
            adir = self._asset_dir(tag)
        except ValueError:
            return False
        return bool(glob.glob(os.path.join(adir, "*.csv")))

    @staticmethod
    def _read_file(path: str) -> pd.DataFrame:
        df = pd.read_csv(path, usecols=["tag", "value", "timestamp"])
        df["timestamp"] = pd.to_datetime(df["timestamp"], utc=True, format="mixed")
        df["value"] = pd.to_numeric(df["value"], errors="coerce")
        return df

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

        files = sorted(
            {
                f
                for tag in tag_list
                for f in glob.glob(os.path.join(self._asset_dir(tag), "*.csv"))
            }
        )
        if not files:
            raise FileNotFoundError(
                f"No IROC csv files for assets of {[t.name for t in tag_list][:5]} "
                f"under {self.base_dir}"
            )
        with ThreadPoolExecutor(max_workers=max(1, self.threads)) as pool:
            frames = list(pool.map(self._read_file, files))
        allrows = pd.concat(frames, ignore_index=True)
        allrows = allrows[
            (allrows["timestamp"] >= from_ts) & (allrows["timestamp"] < to_ts)
        ]
        by_tag = {name: g for name, g in allrows.groupby("tag")}

        wanted = {t.name for t in tag_list}
        extra = set(by_tag) - wanted
        if extra:
            logger.debug("archive holds %d unrequested tags", len(extra))
        for tag in tag_list:
            g = by_tag.get(tag.name)
            if g is None or g.empty:
                # reference behavior: warn about requested tags with no rows
                # and yield an empty series (the dataset join then surfaces it)
                logger.warning("tag %s has no rows in the IROC archive", tag.name)
                yield pd.Series(
                    [], index=pd.DatetimeIndex([], tz="UTC"), name=tag.name, dtype=float
                )
                continue
            s = pd.Series(
                g["value"].values,
                index=pd.DatetimeIndex(g["timestamp"]),
                name=tag.name,
            ).sort_index()
            yield s.iloc[:1] if dry_run else s
