"""Model-layer utilities.

Parity: gordo_components/model/utils.py — ``make_base_dataframe`` (the
multi-level-column output frame shared by server and client),
``metric_wrapper`` (scale y before scoring), and the rolling-window batcher
(the reference's ``create_keras_timeseriesgenerator``; here a zero-copy
strided-view windower — the GPU path has its own gather kernel, K6).
"""
from __future__ import annotations

from typing import Callable, List, Optional, Union

import numpy as np
import pandas as pd


def make_base_dataframe(
    tags: List,
    model_input: np.ndarray,
    model_output: np.ndarray,
    target_tag_list: Optional[List] = None,
    index: Optional[np.ndarray] = None,
    frequency: Optional[pd.Timedelta] = None,
) -> pd.DataFrame:
    """Build the canonical prediction frame with multi-level columns
    ``model-input`` / ``model-output`` (+ ``start``/``end`` when the index is
    time-like)."""

    tag_names = [getattr(t, "name", t) for t in tags]
    target_names = (
        [getattr(t, "name", t) for t in target_tag_list] if target_tag_list else tag_names
    )
    # model output may be longer-featured or shorter-row'd (lookback models)
    offset = model_input.shape[0] - model_output.shape[0]
    model_input = model_input[offset:]
    if index is not None:
        index = np.asarray(index)[offset:]

    names: List[str] = []
    arrays: List[np.ndarray] = []

    names.append("model-input")
    arrays.append(model_input)
    names.append("model-output")
    arrays.append(model_output)

    columns = []
    data_parts = []
    for name, arr in zip(names, arrays):
        sub_names = tag_names if name == "model-input" else target_names
        if arr.shape[1] != len(sub_names):
            sub_names = [str(i) for i in range(arr.shape[1])]
        columns.extend([(name, sn) for sn in sub_names])
        data_parts.append(arr)

    data = np.hstack(data_parts)
    df_index: Union[pd.Index, np.ndarray]
    if index is not None:
        df_index = pd.Index(index)
    else:
        df_index = pd.RangeIndex(len(data))

    df = pd.DataFrame(data, columns=pd.MultiIndex.from_tuples(columns), index=df_index)

    if index is not None and isinstance(df.index, pd.DatetimeIndex) and frequency is not None:
        df[("start", "")] = df.index
        df[("end", "")] = df.index + frequency
    return df


def metric_wrapper(metric: Callable, scaler=None, col: int = None) -> Callable:
    """Wrap an sklearn metric so y/y_pred are transformed by ``scaler`` first
    (parity: the reference scales targets before computing CV metrics so
    per-tag magnitudes don't dominate). ``col`` restricts the metric to one
    tag column (the reference's per-tag CV scores)."""

    def _wrapped(y_true, y_pred, *args, **kwargs):
        y_true = np.asarray(y_true)
        y_pred = np.asarray(y_pred)
        # lookback models emit fewer rows than y; align to the window ends
        if len(y_true) > len(y_pred):
            y_true = y_true[len(y_true) - len(y_pred) :]
        elif len(y_pred) > len(y_true):
            y_pred = y_pred[: len(y_true)]
        if scaler is not None:
            y_true = scaler.transform(y_true)
            y_pred = scaler.transform(y_pred)
        if col is not None and y_true.ndim == 2 and y_true.shape[1] > col:
            y_true = y_true[:, col]
            y_pred = y_pred[:, col]
        return metric(y_true, y_pred, *args, **kwargs)

    _wrapped.__name__ = getattr(metric, "__name__", "metric")
    return _wrapped


def rolling_windows(X: np.ndarray, lookback: int, lookahead: int = 0):
    """(windows, targets) for LSTM training.

    windows[i] = X[i : i+lookback]; target[i] = X[i+lookback-1+lookahead].
    Zero-copy strided view over X (the GPU path gathers on device instead).
    """
    X = np.ascontiguousarray(X)
    n, f = X.shape
    n_windows = n - lookback + 1 - lookahead
    if n_windows <= 0:
        raise ValueError(
            f"Need at least {lookback + lookahead} rows for lookback={lookback} "
            f"lookahead={lookahead}; got {n}"
        )
    s0, s1 = X.strides
    windows = np.lib.stride_tricks.as_strided(
        X, shape=(n_windows, lookback, f), strides=(s0, s0, s1), writeable=False
    )
    targets = X[lookback - 1 + lookahead :]
    return windows, targets
