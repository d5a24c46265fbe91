"""Server (de)serialization helpers.

Parity: gordo_components/server/utils.py — multi-level-column DataFrames go
over the wire either as nested JSON ({"data": {top: {sub: [...]}}}) or as
parquet (pyarrow) with flattened two-level column names.
"""
from __future__ import annotations

import io
from typing import Any, Dict

import numpy as np
import pandas as pd


def dataframe_to_dict(df: pd.DataFrame) -> Dict[str, Any]:
    """Multi-level-column frame → nested JSON-safe dict."""
    out: Dict[str, Any] = {}
    if isinstance(df.columns, pd.MultiIndex):
        for top in df.columns.get_level_values(0).unique():
            sub = df[top]
            if isinstance(sub, pd.Series):
                out[top] = _tolist(sub)
            else:
                out[top] = {str(c): _tolist(sub[c]) for c in sub.columns}
    else:
        for c in df.columns:
            out[str(c)] = _tolist(df[c])
    result = {"data": out}
    if isinstance(df.index, pd.DatetimeIndex):
        result["index"] = [ts.isoformat() for ts in df.index]
    else:
        result["index"] = [int(i) for i in df.index]
    return result


def _tolist(s: pd.Series):
    vals = s.values
    if np.issubdtype(vals.dtype, np.datetime64) or isinstance(s.dtype, pd.DatetimeTZDtype):
        return [pd.Timestamp(v).isoformat() for v in vals]
    return [None if (isinstance(v, float) and np.isnan(v)) else float(v) for v in vals]


def dataframe_from_dict(payload: Dict[str, Any]) -> pd.DataFrame:
    """Inverse of dataframe_to_dict; also accepts plain records / value-lists
    (the reference accepted multiple JSON shapes for X)."""
    data = payload.get("data", payload)
    index = payload.get("index")
    if isinstance(data, list):
        df = pd.DataFrame.from_records(data)
    elif isinstance(data, dict):
        if any(isinstance(v, dict) for v in data.values()):
            # two-level (possibly mixed with flat series like total-anomaly-score)
            frames = {}
            for top, subs in data.items():
                if isinstance(subs, dict):
                    for sub, vals in subs.items():
                        frames[(top, sub)] = vals
                else:
                    frames[(top, "")] = subs
            df = pd.DataFrame(frames)
            df.columns = pd.MultiIndex.from_tuples(df.columns)
        else:
            df = pd.DataFrame(data)
    else:
        raise ValueError(f"Cannot build DataFrame from {type(data)}")
    if index is not None and len(index) == len(df):
        try:
            df.index = pd.DatetimeIndex([pd.Timestamp(i) for i in index])
        except (ValueError, TypeError):
            df.index = index
    return df


# -- parquet ---------------------------------------------------------------

_SEP = "|||"


def dataframe_into_parquet_bytes(df: pd.DataFrame) -> bytes:
    df = df.copy()
    if isinstance(df.columns, pd.MultiIndex):
        df.columns = [_SEP.join(str(p) for p in col).rstrip(_SEP) for col in df.columns]
    else:
        df.columns = [str(c) for c in df.columns]
    buf = io.BytesIO()
    df.to_parquet(buf, engine="pyarrow")
    return buf.getvalue()


def dataframe_from_parquet_bytes(data: bytes) -> pd.DataFrame:
    df = pd.read_parquet(io.BytesIO(data), engine="pyarrow")
    if any(_SEP in str(c) for c in df.columns):
        df.columns = pd.MultiIndex.from_tuples(
            tuple((str(c).split(_SEP) + [""])[:2]) for c in df.columns
        )
    return df
