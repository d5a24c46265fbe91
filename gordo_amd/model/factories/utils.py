"""Shared dim calculators for hourglass-shaped factories.

Parity: gordo_components/model/factories/utils.py — ``hourglass_calc_dims``
(linear interpolation n_features → n_features*compression over
``encoding_layers`` steps, monotonically shrinking, min 1) and
``check_dim_func_len``.
"""
from __future__ import annotations

from typing import List, Optional, Sequence


def hourglass_calc_dims(
    compression_factor: float, encoding_layers: int, n_features: int
) -> List[int]:
    if not (0 <= compression_factor <= 1):
        raise ValueError("compression_factor must be 0<=x<=1")
    if encoding_layers < 1:
        raise ValueError("encoding_layers must be >= 1")
    smallest = max(1, round(n_features * compression_factor))
    dims = [
        max(
            1,
            round(n_features - (n_features - smallest) * (i / encoding_layers)),
        )
        for i in range(1, encoding_layers + 1)
    ]
    return dims


def check_dim_func_len(prefix: str, dim: Sequence[int], func: Optional[Sequence[str]]):
    if func is not None and len(func) != len(dim):
        raise ValueError(
            f"{prefix}_func (len {len(func)}) must match {prefix}_dim (len {len(dim)})"
        )
