"""Anomaly-detector contract.

Parity: gordo_components/model/anomaly/base.py — an AnomalyDetectorBase is a
GordoBase estimator that additionally produces an anomaly frame from
(X, y) after fitting.
"""
from __future__ import annotations

import pandas as pd

from ..base import GordoBase


class AnomalyDetectorBase(GordoBase):
    def anomaly(self, X: pd.DataFrame, y: pd.DataFrame, frequency=None) -> pd.DataFrame:
        """Return a frame with at least ``tag-anomaly-scores`` and
        ``total-anomaly-score`` column groups."""
        raise NotImplementedError()
