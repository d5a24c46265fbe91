"""Dataset implementations.

Parity: gordo_components/dataset/datasets.py — TimeSeriesDataset pulls
per-tag series from a provider, resamples each to ``resolution`` with the
configured aggregation method(s), outer-joins on timestamp, forward-fills
within limits, applies ``row_filter``, and returns ``(X, y)`` DataFrames
(y = the target-tag subset, or None so autoencoders train X→X).
RandomDataset is TimeSeriesDataset pre-wired to RandomDataProvider.
"""
from __future__ import annotations

import logging
from datetime import datetime
from typing import Any, Dict, Iterable, List, Optional, Tuple, Union

import pandas as pd

from .base import GordoBaseDataset, InsufficientDataError
from .data_provider.base import GordoBaseDataProvider
from .data_provider.providers import RandomDataProvider
from .filter_rows import pandas_filter_rows
from ..machine.sensor_tag import normalize_sensor_tags, to_list_of_strings

logger = logging.getLogger(__name__)


def _to_ts(value: Union[str, datetime]) -> pd.Timestamp:
    ts = pd.Timestamp(value)
    if ts.tzinfo is None:
        ts = ts.tz_localize("UTC")
    return ts


class TimeSeriesDataset(GordoBaseDataset):
    def __init__(
        self,
        train_start_date: Union[str, datetime],
        train_end_date: Union[str, datetime],
        tag_list: Iterable,
        target_tag_list: Optional[Iterable] = None,
        data_provider: Union[GordoBaseDataProvider, dict, None] = None,
        resolution: str = "10min",
        row_filter: Union[str, list] = "",
        known_filter_periods: Optional[list] = None,
        aggregation_methods: Union[str, List[str]] = "mean",
        row_filter_buffer_size: int = 0,
        n_samples_threshold: int = 0,
        asset: Optional[str] = None,
        interpolation_method: str = "linear_interpolation",
        interpolation_limit: str = "8h",
    ):
        self.train_start_date = _to_ts(train_start_date)
        self.train_end_date = _to_ts(train_end_date)
        if self.train_start_date >= self.train_end_date:
            raise ValueError(
                f"train_start_date ({self.train_start_date}) must be before "
                f"train_end_date ({self.train_end_date})"
            )
        self.tag_list = normalize_sensor_tags(list(tag_list), asset)
        self.target_tag_list = (
            normalize_sensor_tags(list(target_tag_list), asset)
            if target_tag_list
            else []
        )
        if isinstance(data_provider, dict):
            data_provider = GordoBaseDataProvider.from_dict(data_provider)
        self.data_provider = data_provider or RandomDataProvider()
        self.resolution = resolution
        self.row_filter = row_filter
        self.known_filter_periods = known_filter_periods or []
        self.aggregation_methods = aggregation_methods
        self.row_filter_buffer_size = row_filter_buffer_size
        self.n_samples_threshold = n_samples_threshold
        self.asset = asset
        self.interpolation_method = interpolation_method
        self.interpolation_limit = interpolation_limit
        self._metadata: Dict[str, Any] = {}

    # -- assembly ---------------------------------------------------------

    def join_timeseries(
        self,
        series_iterable: Iterable[pd.Series],
        resampling_startpoint: pd.Timestamp,
        resampling_endpoint: pd.Timestamp,
        resolution: str,
        aggregation_methods: Union[str, List[str]] = "mean",
        interpolation_method: str = "linear_interpolation",
        interpolation_limit: str = "8h",
    ) -> pd.DataFrame:
        """Resample each series to the common grid, then outer-join."""
        resampled: List[pd.DataFrame] = []
        for series in series_iterable:
            if series.empty:
                raise InsufficientDataError(f"Tag {series.name!r} returned no data")
            startpoint_sametz = resampling_startpoint.tz_convert(series.index.tz)
            endpoint_sametz = resampling_endpoint.tz_convert(series.index.tz)
            # anchor the resampling grid by padding the edges (parity with the
            # reference, which inserts NaN rows at the exact boundaries)
            pad = pd.Series(
                [None, None],
                index=[startpoint_sametz, endpoint_sametz],
                dtype="float64",
                name=series.name,
            )
            series = pd.concat([series, pad]).sort_index()
            resampler = series.resample(resolution, origin=startpoint_sametz)
            agg = resampler.agg(aggregation_methods)

            limit = max(1, int(pd.Timedelta(interpolation_limit) / pd.Timedelta(resolution)))
            if interpolation_method == "linear_interpolation":
                agg = agg.interpolate(method="linear", limit=limit, limit_direction="forward")
            else:  # ffill
                agg = agg.ffill(limit=limit)
            frame = agg.to_frame(name=series.name) if isinstance(agg, pd.Series) else agg
            if isinstance(agg, pd.DataFrame):
                # multiple aggregation methods → multi-level columns
                frame = agg
                frame.columns = pd.MultiIndex.from_product([[series.name], frame.columns])
            resampled.append(frame)
        joined = pd.concat(resampled, axis=1, join="outer")
        return joined.dropna()

    def get_data(self) -> Tuple[pd.DataFrame, Optional[pd.DataFrame]]:
        tags = list(
            dict.fromkeys(self.tag_list + self.target_tag_list)
        )  # ordered de-dup
        series = self.data_provider.load_series(
            self.train_start_date.to_pydatetime(),
            self.train_end_date.to_pydatetime(),
            tags,
        )
        data = self.join_timeseries(
            series,
            self.train_start_date,
            self.train_end_date,
            self.resolution,
            aggregation_methods=self.aggregation_methods,
            interpolation_method=self.interpolation_method,
            interpolation_limit=self.interpolation_limit,
        )

        for period in self.known_filter_periods:
            if period:
                data = pandas_filter_rows(data, period, buffer_size=0)

        if self.row_filter:
            data = pandas_filter_rows(
                data, self.row_filter, buffer_size=self.row_filter_buffer_size
            )

        if len(data) <= self.n_samples_threshold:
            raise InsufficientDataError(
                f"Only {len(data)} rows after filtering "
                f"(threshold {self.n_samples_threshold})"
            )

        # column order follows the CONFIGURED tag order (not join order):
        # downstream naming — model outputs, thresholds, anomaly columns —
        # assumes X matches tag_list and y matches target_tag_list
        def _ordered(names: List[str]) -> List:
            return [c for n in names for c in data.columns if _col_tag(c) == n]

        X = data[_ordered(to_list_of_strings(self.tag_list))]
        y = None
        if self.target_tag_list:
            y = data[_ordered(to_list_of_strings(self.target_tag_list))]

        self._metadata = {
            "tag_list": [t.to_json() for t in self.tag_list],
            "target_tag_list": [t.to_json() for t in self.target_tag_list],
            "train_start_date": self.train_start_date.isoformat(),
            "train_end_date": self.train_end_date.isoformat(),
            "resolution": self.resolution,
            "row_filter": self.row_filter,
            "dataset_samples": len(X),
            "x_columns": [str(c) for c in X.columns],
        }
        return X, y

    def get_metadata(self) -> Dict[str, Any]:
        return dict(self._metadata)

    def to_dict(self) -> Dict[str, Any]:
        out = super().to_dict()
        out["train_start_date"] = self.train_start_date.isoformat()
        out["train_end_date"] = self.train_end_date.isoformat()
        out["tag_list"] = [t.to_json() for t in self.tag_list]
        out["target_tag_list"] = [t.to_json() for t in self.target_tag_list]
        out["data_provider"] = self.data_provider.to_dict()
        out.pop("asset", None)
        return out


def _col_tag(col) -> str:
    """Column → tag name (first level when multi-agg produced MultiIndex)."""
    if isinstance(col, tuple):
        return col[0]
    return col


class RandomDataset(TimeSeriesDataset):
    """TimeSeriesDataset pre-wired to synthetic data (parity: RandomDataset)."""

    def __init__(self, train_start_date, train_end_date, tag_list, **kwargs):
        kwargs.pop("data_provider", None)
        min_size = kwargs.pop("min_size", 100)
        max_size = kwargs.pop("max_size", 300)
        super().__init__(
            train_start_date,
            train_end_date,
            tag_list,
            data_provider=RandomDataProvider(min_size=min_size, max_size=max_size),
            **kwargs,
        )
