"""NCS/IROC-shaped reader tests against the checked-in fixture trees
(tests/data/ncs_tree, tests/data/iroc_tree — see make_fixtures.py).

Reader-semantics parity targets (VERDICT r1 missing #3):
per-tag per-year file layout, asset-prefix routing, threadpooled reads,
misaligned years, missing tags, tz handling.
Reference: gordo_components/dataset/data_provider/ncs_reader.py,
iroc_reader.py.
"""
import datetime
import os

import numpy as np
import pandas as pd
import pytest

from gordo_amd.dataset.data_provider import IrocFileProvider, NcsFileProvider
from gordo_amd.dataset.data_provider.base import GordoBaseDataProvider
from gordo_amd.machine.sensor_tag import SensorTag, normalize_sensor_tags

HERE = os.path.dirname(os.path.abspath(__file__))
NCS = os.path.join(HERE, "data", "ncs_tree")
IROC = os.path.join(HERE, "data", "iroc_tree")

UTC = datetime.timezone.utc
FROM = datetime.datetime(2019, 12, 30, tzinfo=UTC)
TO = datetime.datetime(2020, 1, 10, tzinfo=UTC)


def test_ncs_reads_across_years_and_assets():
    """TAG-A spans a 2019 and a 2020 parquet; TAG-C lives under a different
    asset dir routed purely from the tag-name prefix."""
    p = NcsFileProvider(NCS)
    tags = normalize_sensor_tags(["ASGB.TAG-A", "GRA.TAG-C"])
    series = list(p.load_series(FROM, TO, tags))
    assert [s.name for s in series] == ["ASGB.TAG-A", "GRA.TAG-C"]
    a, c = series
    # rows from both year files survive, window-filtered [from, to)
    assert a.index.min() >= pd.Timestamp(FROM)
    assert a.index.max() < pd.Timestamp(TO)
    assert a.index.min().year == 2019 and a.index.max().year == 2020
    assert a.index.is_monotonic_increasing
    # parquet preferred over the decoy csv for the same (tag, year)
    assert not (a == -999.0).any()
    # naive CSV timestamps read as UTC
    assert str(c.index.tz) == "UTC"
    assert len(c) > 0


def test_ncs_misaligned_years_tolerated():
    """TAG-B only has a 2020 file; the missing 2019 file is skipped, not
    fatal (ragged archives are the reference's normal case)."""
    p = NcsFileProvider(NCS)
    (b,) = list(p.load_series(FROM, TO, normalize_sensor_tags(["ASGB.TAG-B"])))
    assert len(b) > 0
    assert b.index.min().year == 2020


def test_ncs_totally_missing_tag_raises():
    p = NcsFileProvider(NCS)
    with pytest.raises(FileNotFoundError, match="NOSUCH"):
        list(p.load_series(FROM, TO, [SensorTag("ASGB.NOSUCH", "1191-asgb")]))


def test_ncs_unroutable_tag_raises():
    p = NcsFileProvider(NCS)
    with pytest.raises(ValueError, match="route"):
        list(p.load_series(FROM, TO, [SensorTag("UNKNOWN.PREFIX-1", None)]))


def test_ncs_can_handle_tag():
    p = NcsFileProvider(NCS)
    assert p.can_handle_tag(SensorTag("ASGB.TAG-A", "1191-asgb"))
    assert p.can_handle_tag(SensorTag("GRA.TAG-C", None))  # prefix-routed
    assert not p.can_handle_tag(SensorTag("ASGB.NOSUCH", "1191-asgb"))
    assert not p.can_handle_tag(SensorTag("NOPREFIX", None))


def test_ncs_csv_semicolon_layout_and_values():
    """CSV years use the reference's headerless Sensor;Value;Timestamp rows."""
    p = NcsFileProvider(NCS, lookup_types=["csv"])
    (b,) = list(p.load_series(FROM, TO, normalize_sensor_tags(["ASGB.TAG-B"])))
    raw = pd.read_csv(
        os.path.join(NCS, "1191-asgb", "ASGB.TAG-B", "ASGB.TAG-B_2020.csv"),
        sep=";",
        header=None,
        names=["Sensor", "Value", "Timestamp"],
    )
    assert len(b) <= len(raw)
    np.testing.assert_allclose(b.values, raw["Value"].values[: len(b)])


def test_ncs_threadpool_width_does_not_change_result():
    tags = normalize_sensor_tags(["ASGB.TAG-A", "ASGB.TAG-B", "GRA.TAG-C"])
    wide = list(NcsFileProvider(NCS, threads=8).load_series(FROM, TO, tags))
    narrow = list(NcsFileProvider(NCS, threads=1).load_series(FROM, TO, tags))
    for s1, s2 in zip(wide, narrow):
        pd.testing.assert_series_equal(s1, s2)


def test_ncs_dry_run_single_row():
    p = NcsFileProvider(NCS)
    series = list(
        p.load_series(FROM, TO, normalize_sensor_tags(["ASGB.TAG-A"]), dry_run=True)
    )
    assert len(series[0]) == 1


def test_ncs_from_dict_short_name():
    p = GordoBaseDataProvider.from_dict({"type": "NcsFileProvider", "base_dir": NCS})
    assert isinstance(p, NcsFileProvider)
    rt = p.to_dict()
    assert rt["base_dir"] == NCS


# ---------------------------------------------------------------------------
# IROC
# ---------------------------------------------------------------------------


def test_iroc_batched_csvs_pivot_to_series():
    p = IrocFileProvider(IROC)
    tags = normalize_sensor_tags(["ASGB.IROC-X", "ASGB.IROC-Y"])
    x, y = list(p.load_series(FROM, TO, tags))
    assert x.name == "ASGB.IROC-X" and y.name == "ASGB.IROC-Y"
    # rows from BOTH batch files, merged + sorted
    assert len(x) == 96 and len(y) == 96
    assert x.index.is_monotonic_increasing
    assert str(x.index.tz) == "UTC"


def test_iroc_missing_tag_yields_empty_series_with_warning(caplog):
    p = IrocFileProvider(IROC)
    with caplog.at_level("WARNING"):
        x, ghost = list(
            p.load_series(
                FROM, TO, normalize_sensor_tags(["ASGB.IROC-X", "ASGB.GHOST"])
            )
        )
    assert len(x) == 96
    assert len(ghost) == 0 and ghost.name == "ASGB.GHOST"
    assert any("ASGB.GHOST" in r.message for r in caplog.records)


def test_iroc_window_filter():
    p = IrocFileProvider(IROC)
    to = datetime.datetime(2020, 1, 2, tzinfo=UTC)
    (x,) = list(p.load_series(FROM, to, normalize_sensor_tags(["ASGB.IROC-X"])))
    assert len(x) == 24  # only batch 1's first day
    assert x.index.max() < pd.Timestamp(to)


def test_iroc_no_files_raises():
    p = IrocFileProvider(os.path.join(HERE, "data", "ncs_tree"))  # wrong tree
    with pytest.raises(FileNotFoundError):
        list(p.load_series(FROM, TO, normalize_sensor_tags(["ASGB.IROC-X"])))


def test_iroc_dataset_end_to_end():
    """The NCS provider slots into TimeSeriesDataset like any other."""
    from gordo_amd.dataset.datasets import TimeSeriesDataset

    ds = TimeSeriesDataset(
        train_start_date=FROM,
        train_end_date=TO,
        tag_list=["ASGB.TAG-A", "ASGB.TAG-B"],
        resolution="6h",
        data_provider={"type": "NcsFileProvider", "base_dir": NCS},
    )
    X, y = ds.get_data()
    assert list(X.columns) == ["ASGB.TAG-A", "ASGB.TAG-B"]
    assert len(X) > 0
    assert not X.isna().any().any()
