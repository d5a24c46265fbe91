"""Dataset & provider tests (parity: tests/.../test_dataset.py)."""
import os
from datetime import datetime, timezone
from pathlib import Path

import numpy as np
import pandas as pd
import pytest

from gordo_amd.dataset import (
    GordoBaseDataset,
    InsufficientDataError,
    RandomDataset,
    TimeSeriesDataset,
    dataset_from_config,
)
from gordo_amd.dataset.data_provider import (
    CsvFileProvider,
    ParquetFileProvider,
    RandomDataProvider,
)
from gordo_amd.dataset.filter_rows import pandas_filter_rows, parse_pandas_filter_vars
from gordo_amd.machine.sensor_tag import (
    SensorTag,
    normalize_sensor_tag,
    normalize_sensor_tags,
)

_REPO_ROOT = Path(__file__).resolve().parent.parent

TAGS = [f"tag-{i}" for i in range(5)]
START, END = "2020-01-01T00:00:00Z", "2020-02-01T00:00:00Z"


def test_random_dataset_get_data():
    ds = RandomDataset(START, END, TAGS)
    X, y = ds.get_data()
    assert isinstance(X, pd.DataFrame)
    assert list(X.columns) == TAGS
    assert y is None
    assert len(X) > 10
    meta = ds.get_metadata()
    assert meta["dataset_samples"] == len(X)


def test_target_tag_list_gives_y():
    ds = RandomDataset(START, END, TAGS, target_tag_list=TAGS[:2])
    X, y = ds.get_data()
    assert y is not None
    assert list(y.columns) == TAGS[:2]
    assert len(X) == len(y)


def test_dataset_from_config_roundtrip():
    ds = RandomDataset(START, END, TAGS)
    config = ds.to_dict()
    ds2 = dataset_from_config(config)
    assert isinstance(ds2, RandomDataset)
    X1, _ = ds.get_data()
    X2, _ = ds2.get_data()
    pd.testing.assert_frame_equal(X1, X2)


def test_random_provider_deterministic_across_processes():
    """Seeds must not depend on Python's per-process hash salt: a fresh
    interpreter (different PYTHONHASHSEED) must produce identical data."""
    import subprocess
    import sys

    script = (
        "import sys; sys.path.insert(0, %r)\n"
        "from gordo_amd.dataset import RandomDataset\n"
        "X, _ = RandomDataset(%r, %r, %r).get_data()\n"
        "print(X.to_numpy().sum(), X.shape)\n" % (str(_REPO_ROOT), START, END, TAGS)
    )
    outs = [
        subprocess.run(
            [sys.executable, "-c", script],
            capture_output=True, text=True, check=True,
            env={**os.environ, "PYTHONHASHSEED": seed},
        ).stdout
        for seed in ("1", "2")
    ]
    assert outs[0] == outs[1] and outs[0].strip()


def test_invalid_date_order_raises():
    with pytest.raises(ValueError):
        RandomDataset(END, START, TAGS)


def test_insufficient_data_threshold():
    ds = RandomDataset(
        "2020-01-01T00:00:00Z", "2020-01-01T04:00:00Z", TAGS, n_samples_threshold=10**6
    )
    with pytest.raises(InsufficientDataError):
        ds.get_data()


def test_row_filter():
    ds = RandomDataset(START, END, TAGS, row_filter=f"`tag-0` > -1e9")
    X, _ = ds.get_data()
    assert len(X) > 0
    ds2 = RandomDataset(START, END, TAGS, row_filter=f"`tag-0` > 1e9")
    with pytest.raises(InsufficientDataError):
        ds2.get_data()


def test_join_timeseries_misaligned():
    idx_a = pd.date_range("2020-01-01", periods=50, freq="7min", tz="UTC")
    idx_b = pd.date_range("2020-01-01", periods=80, freq="5min", tz="UTC")
    series = [
        pd.Series(np.arange(50, dtype=float), index=idx_a, name="a"),
        pd.Series(np.arange(80, dtype=float), index=idx_b, name="b"),
    ]
    ds = TimeSeriesDataset(START, END, ["a", "b"], data_provider=RandomDataProvider())
    joined = ds.join_timeseries(
        iter(series),
        pd.Timestamp("2020-01-01", tz="UTC"),
        pd.Timestamp("2020-01-02", tz="UTC"),
        "10min",
    )
    assert set(joined.columns) == {"a", "b"}
    assert joined.notna().all().all()


def test_pandas_filter_rows():
    df = pd.DataFrame({"a": [0.1, 0.9, 0.5], "b": [1.0, 2.0, 3.0]})
    out = pandas_filter_rows(df, "a > 0.3")
    assert len(out) == 2
    out = pandas_filter_rows(df, ["a > 0.3", "b < 2.5"])
    assert len(out) == 1
    with pytest.raises(ValueError):
        pandas_filter_rows(df, "__import__('os').system('true')")


def test_parse_filter_vars():
    assert "tag one" in parse_pandas_filter_vars("`tag one` > 0.5")
    assert "simple_tag" in parse_pandas_filter_vars("simple_tag > 0.5 & other < 1")


def test_sensor_tag_normalization():
    assert normalize_sensor_tag("ASGB.A.B") == SensorTag("ASGB.A.B", "1191-asgb")
    assert normalize_sensor_tag(["x", "assetx"]) == SensorTag("x", "assetx")
    assert normalize_sensor_tag({"name": "x", "asset": "a"}) == SensorTag("x", "a")
    tags = normalize_sensor_tags(["a", "b"], asset="z")
    assert all(t.asset == "z" for t in tags)


def test_file_providers(tmp_path):
    idx = pd.date_range("2020-01-01", periods=100, freq="10min", tz="UTC")
    for name in ("t1", "t2"):
        df = pd.DataFrame({"timestamp": idx, "value": np.arange(100, dtype=float)})
        df.to_parquet(tmp_path / f"{name}.parquet")
        df.to_csv(tmp_path / f"{name}.csv", index=False)
    for provider in (
        ParquetFileProvider(str(tmp_path)),
        CsvFileProvider(str(tmp_path)),
    ):
        series = list(
            provider.load_series(
                datetime(2020, 1, 1, tzinfo=timezone.utc),
                datetime(2020, 1, 2, tzinfo=timezone.utc),
                [SensorTag("t1"), SensorTag("t2")],
            )
        )
        assert len(series) == 2
        assert all(len(s) > 0 for s in series)
        assert provider.can_handle_tag(SensorTag("t1"))
        assert not provider.can_handle_tag(SensorTag("missing"))


def test_dataset_with_file_provider(tmp_path):
    idx = pd.date_range("2020-01-01", periods=200, freq="10min", tz="UTC")
    rng = np.random.default_rng(0)
    for name in TAGS:
        df = pd.DataFrame({"timestamp": idx, "value": rng.normal(size=200)})
        df.to_parquet(tmp_path / f"{name}.parquet")
    ds = TimeSeriesDataset(
        "2020-01-01T00:00:00Z",
        "2020-01-02T00:00:00Z",
        TAGS,
        data_provider={"type": "ParquetFileProvider", "base_dir": str(tmp_path)},
    )
    X, _ = ds.get_data()
    assert list(X.columns) == TAGS
    assert len(X) > 50


def test_multiple_aggregation_methods():
    """aggregation_methods as a list yields (tag, method) MultiIndex columns
    (reference TimeSeriesDataset behavior)."""
    from gordo_amd.dataset.datasets import TimeSeriesDataset

    ds = TimeSeriesDataset(
        train_start_date="2020-01-01T00:00:00Z",
        train_end_date="2020-01-03T00:00:00Z",
        tag_list=["tag-0", "tag-1"],
        data_provider={"type": "RandomDataProvider"},
        aggregation_methods=["mean", "max"],
    )
    X, _ = ds.get_data()
    assert isinstance(X.columns, pd.MultiIndex)
    assert ("tag-0", "mean") in X.columns and ("tag-1", "max") in X.columns
    assert X.shape[1] == 4
    assert (X["tag-0"]["max"] >= X["tag-0"]["mean"] - 1e-9).all()


def test_target_tag_order_preserved():
    """y columns must follow target_tag_list ORDER (not the join order):
    downstream output naming assumes it."""
    ds = RandomDataset(
        train_start_date="2020-01-01T00:00:00Z",
        train_end_date="2020-02-01T00:00:00Z",
        tag_list=["tag-a", "tag-b", "tag-c"],
        target_tag_list=["tag-c", "tag-a"],
    )
    X, y = ds.get_data()
    assert list(X.columns) == ["tag-a", "tag-b", "tag-c"]
    assert list(y.columns) == ["tag-c", "tag-a"]
