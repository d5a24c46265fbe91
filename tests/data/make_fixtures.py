#!/usr/bin/env python3
"""Regenerates the checked-in NCS/IROC fixture trees under tests/data/.

Run from the repo root:  python tests/data/make_fixtures.py

Layouts (reader-semantics parity with the reference's ncs_reader/iroc_reader):

ncs_tree/                       one file per tag per YEAR, per-asset dirs
  1191-asgb/
    ASGB.TAG-A/ASGB.TAG-A_2019.parquet   (Timestamp, Value columns)
    ASGB.TAG-A/ASGB.TAG-A_2020.parquet
    ASGB.TAG-A/ASGB.TAG-A_2020.csv       (parquet preferred when both exist)
    ASGB.TAG-B/ASGB.TAG-B_2020.csv       headerless `Sensor;Value;Timestamp`
                                         (misaligned years: no 2019 file)
  1755-gra/
    GRA.TAG-C/GRA.TAG-C_2019.csv
    GRA.TAG-C/GRA.TAG-C_2020.csv         naive timestamps -> read as UTC

iroc_tree/                      batched CSVs: many tags per file
  1191-asgb/batch_2020_01.csv   header tag,value,timestamp
  1191-asgb/batch_2020_02.csv
"""
import os

import numpy as np
import pandas as pd

HERE = os.path.dirname(os.path.abspath(__file__))


def hours(start, n, tz="UTC"):
    return pd.date_range(start, periods=n, freq="1h", tz=tz)


def tag_values(seed, n):
    return np.round(np.cumsum(np.random.default_rng(seed).normal(0, 0.1, n)) + 10.0, 4)


def write_ncs():
    root = os.path.join(HERE, "ncs_tree")

    def pq(tag, year, idx, vals):
        d = os.path.join(root, asset_of(tag), tag)
        os.makedirs(d, exist_ok=True)
        pd.DataFrame({"Timestamp": idx, "Value": vals}).to_parquet(
            os.path.join(d, f"{tag}_{year}.parquet"), index=False
        )

    def csv(tag, year, idx, vals, naive=False):
        d = os.path.join(root, asset_of(tag), tag)
        os.makedirs(d, exist_ok=True)
        stamps = idx.tz_localize(None) if naive else idx
        with open(os.path.join(d, f"{tag}_{year}.csv"), "w") as f:
            for t, v in zip(stamps, vals):
                f.write(f"{tag};{v};{t.isoformat()}\n")

    def asset_of(tag):
        return {"ASGB": "1191-asgb", "GRA": "1755-gra"}[tag.split(".")[0]]

    # ASGB.TAG-A: parquet for 2019 + 2020, plus a DIFFERENT 2020 csv that must
    # be shadowed by the parquet (lookup order)
    i19 = hours("2019-12-30", 72)
    pq("ASGB.TAG-A", 2019, i19, tag_values(1, 72))
    i20 = hours("2020-01-02", 96)
    pq("ASGB.TAG-A", 2020, i20, tag_values(2, 96))
    csv("ASGB.TAG-A", 2020, i20, np.full(96, -999.0))

    # ASGB.TAG-B: csv only, 2020 only (misaligned years vs TAG-A)
    ib = hours("2020-01-01", 120)
    csv("ASGB.TAG-B", 2020, ib, tag_values(3, 120))

    # GRA.TAG-C: csv with NAIVE timestamps (tz handling: read as UTC)
    ic19 = hours("2019-12-31", 48)
    csv("GRA.TAG-C", 2019, ic19, tag_values(4, 48), naive=True)
    ic20 = hours("2020-01-02", 48)
    csv("GRA.TAG-C", 2020, ic20, tag_values(5, 48), naive=True)


def write_iroc():
    root = os.path.join(HERE, "iroc_tree", "1191-asgb")
    os.makedirs(root, exist_ok=True)
    idx1 = hours("2020-01-01", 48)
    idx2 = hours("2020-01-03", 48)
    for fname, idx, seeds in [
        ("batch_2020_01.csv", idx1, (10, 11)),
        ("batch_2020_02.csv", idx2, (12, 13)),
    ]:
        rows = []
        for tag, seed in zip(["ASGB.IROC-X", "ASGB.IROC-Y"], seeds):
            vals = tag_values(seed, len(idx))
            rows.extend(
                {"tag": tag, "value": v, "timestamp": t.isoformat()}
                for t, v in zip(idx, vals)
            )
        pd.DataFrame(rows).to_csv(os.path.join(root, fname), index=False)


if __name__ == "__main__":
    write_ncs()
    write_iroc()
    print("fixture trees written under", HERE)
