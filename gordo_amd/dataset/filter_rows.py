"""Row filtering via safe pandas expressions.

Parity: gordo_components/dataset/filter_rows.py — YAML configs may carry a
``row_filter`` boolean expression over tag columns (e.g. ``"`TAG A` > 0.5 &
`TAG B` < 1"``); rows failing the filter are dropped before training.
"""
from __future__ import annotations

import re
from typing import List, Union

import pandas as pd

_ALLOWED = re.compile(
    r"""^[\w\s\.\,\'\"\`\<\>\=\!\&\|\~\+\-\*\/\(\)\[\]\@\:]*$"""
)  # ':' for timestamps in period filters, e.g. "index <= '2020-01-02 06:00:00'"


def parse_pandas_filter_vars(expr: Union[str, List[str]]) -> List[str]:
    """Extract backtick-quoted and bare identifiers referenced by the filter."""
    if isinstance(expr, list):
        out: List[str] = []
        for e in expr:
            out.extend(parse_pandas_filter_vars(e))
        return out
    vars_quoted = re.findall(r"`([^`]+)`", expr)
    stripped = re.sub(r"`[^`]+`", "", expr)
    stripped = re.sub(r"'[^']*'|\"[^\"]*\"", "", stripped)
    bare = re.findall(r"[A-Za-z_][\w\.\-]*", stripped)
    keywords = {"and", "or", "not", "in", "True", "False", "abs", "index"}
    return vars_quoted + [b for b in bare if b not in keywords]


def pandas_filter_rows(
    df: pd.DataFrame, filter_str: Union[str, List[str]], buffer_size: int = 0
) -> pd.DataFrame:
    """Keep rows where the expression evaluates truthy.

    ``filter_str`` may be a single expression or a list (AND-combined).
    ``buffer_size`` also drops the N rows before/after every dropped row
    (parity with the reference's filter buffer).
    """
    if isinstance(filter_str, list):
        filter_str = " & ".join(f"({f})" for f in filter_str)
    if not _ALLOWED.match(filter_str):
        raise ValueError(f"Unsafe characters in row_filter: {filter_str!r}")
    mask = df.eval(filter_str)
    if not pd.api.types.is_bool_dtype(mask):
        raise ValueError(f"row_filter must evaluate to booleans: {filter_str!r}")
    if buffer_size:
        bad = ~mask
        # expand the dropped region by buffer_size rows on each side
        bad = bad.rolling(2 * buffer_size + 1, center=True, min_periods=1).max().astype(bool)
        mask = ~bad
    return df[mask]
