"""Unit tests for the small utility layers: disk registry (the builder's
cache/recovery unit) and the dataset's filter-period/buffer plumbing."""
import numpy as np
import pandas as pd
import pytest

from gordo_amd.util import disk_registry


def test_disk_registry_roundtrip(tmp_path):
    reg = tmp_path / "reg"
    assert disk_registry.get_value(reg, "abc") is None
    disk_registry.write_key(reg, "abc", "/models/m1")
    assert disk_registry.get_value(reg, "abc") == "/models/m1"
    disk_registry.write_key(reg, "abc", "/models/m2")  # overwrite
    assert disk_registry.get_value(reg, "abc") == "/models/m2"
    assert disk_registry.delete_value(reg, "abc") is True
    assert disk_registry.get_value(reg, "abc") is None
    assert disk_registry.delete_value(reg, "abc") is False


def test_disk_registry_sanitizes_keys(tmp_path):
    """Keys with path-hostile characters (config hashes are hex, but be
    safe) never escape the registry dir."""
    reg = tmp_path / "reg"
    disk_registry.write_key(reg, "../evil/key", "v")
    files = list(reg.iterdir())
    assert len(files) == 1
    assert files[0].parent == reg
    assert disk_registry.get_value(reg, "../evil/key") == "v"


def test_known_filter_periods_drop_rows():
    from gordo_amd.dataset.datasets import TimeSeriesDataset

    ds = TimeSeriesDataset(
        train_start_date="2020-01-01T00:00:00Z",
        train_end_date="2020-01-05T00:00:00Z",
        tag_list=["tag-0", "tag-1"],
        data_provider={"type": "RandomDataProvider"},
        known_filter_periods=[
            "~(index >= '2020-01-02 00:00:00+00:00') | ~(index <= '2020-01-03 00:00:00+00:00')"
        ],
    )
    X, _ = ds.get_data()
    masked = (X.index >= "2020-01-02T00:00:00Z") & (X.index <= "2020-01-03T00:00:00Z")
    assert masked.sum() == 0
    assert len(X) > 0


def test_row_filter_buffer_expands_drop_region():
    """row_filter_buffer_size drops N rows on each side of every filtered
    region (the reference's transient-settling semantics)."""
    from gordo_amd.dataset.filter_rows import pandas_filter_rows

    idx = pd.date_range("2021-01-01", periods=20, freq="10min")
    df = pd.DataFrame({"a": np.ones(20)}, index=idx)
    df.loc[idx[10], "a"] = -1.0  # one bad row
    kept = pandas_filter_rows(df, "a > 0", buffer_size=0)
    assert len(kept) == 19
    kept_buf = pandas_filter_rows(df, "a > 0", buffer_size=2)
    assert len(kept_buf) == 15  # bad row + 2 neighbours on each side
    assert idx[8] not in kept_buf.index and idx[12] not in kept_buf.index
    assert idx[7] in kept_buf.index and idx[13] in kept_buf.index
