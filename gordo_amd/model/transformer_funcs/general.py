"""Small sklearn-compatible transformer helpers referenceable from YAML
pipelines (parity: gordo_components/model/transformer_funcs/general.py and
model/transformers/imputer.py)."""
from __future__ import annotations

import numpy as np
from sklearn.base import BaseEstimator, TransformerMixin


def multiply(X, factor: float = 1.0):
    """FunctionTransformer-compatible elementwise multiply."""
    return np.asarray(X) * factor


class InfImputer(BaseEstimator, TransformerMixin):
    """Replace ±inf (and optionally clip extremes) so downstream scalers and
    models never see non-finite values."""

    def __init__(self, inf_fill_value=None, neg_inf_fill_value=None, strategy="extremes", delta: float = 2.0):
        self.inf_fill_value = inf_fill_value
        self.neg_inf_fill_value = neg_inf_fill_value
        self.strategy = strategy
        self.delta = delta

    def fit(self, X, y=None):
        X = np.asarray(X, dtype=np.float64)
        finite = np.where(np.isfinite(X), X, np.nan)
        if self.strategy == "extremes":
            self._pos_fill = np.nanmax(finite, axis=0) + self.delta
            self._neg_fill = np.nanmin(finite, axis=0) - self.delta
        elif self.strategy == "minmax":
            self._pos_fill = np.full(X.shape[1], np.finfo(np.float32).max)
            self._neg_fill = np.full(X.shape[1], np.finfo(np.float32).min)
        else:
            raise ValueError(f"Unknown strategy {self.strategy!r}")
        if self.inf_fill_value is not None:
            self._pos_fill = np.full(X.shape[1], self.inf_fill_value)
        if self.neg_inf_fill_value is not None:
            self._neg_fill = np.full(X.shape[1], self.neg_inf_fill_value)
        return self

    def transform(self, X, y=None):
        X = np.array(X, dtype=np.float64, copy=True)
        pos = np.isposinf(X)
        neg = np.isneginf(X)
        if pos.any():
            X[pos] = np.broadcast_to(self._pos_fill, X.shape)[pos]
        if neg.any():
            X[neg] = np.broadcast_to(self._neg_fill, X.shape)[neg]
        return X
