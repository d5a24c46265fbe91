"""Model runtime for the server.

Parity: gordo_components/server/model_io.py — the model dir is loaded ONCE at
startup. MI355X-native addition: when the loaded object is the canonical
MinMaxScaler→TorchAutoEncoder pipeline inside a DiffBasedAnomalyDetector and a
GPU is present, the whole serving hot path (scale → forward → per-feature
error → row L2) runs as ONE fused HIP kernel (SURVEY.md §3.2 / K1+K7+K8).
"""
from __future__ import annotations

import logging
import os
from typing import Any, Dict, Optional

import numpy as np
import pandas as pd
import torch

from ..model.anomaly.base import AnomalyDetectorBase
from ..serializer import load, load_metadata

logger = logging.getLogger(__name__)


class ModelRuntime:
    def __init__(self, model_dir: str, device: Optional[str] = None):
        self.model_dir = model_dir
        self.device = torch.device(
            device
            or os.environ.get("GORDO_DEVICE")
            or ("cuda:0" if torch.cuda.is_available() else "cpu")
        )
        self.model = load(model_dir)
        try:
            self.metadata: Dict[str, Any] = load_metadata(model_dir)
        except FileNotFoundError:
            self.metadata = {}
        self.is_anomaly = isinstance(self.model, AnomalyDetectorBase)
        self._fused = None
        self._maybe_build_fused()

    # ------------------------------------------------------------------
    def _maybe_build_fused(self):
        if not torch.cuda.is_available():
            return
        try:
            from ..ops import is_available
            from ..ops.hourglass_engine import FusedHourglassForward, supported

            if not is_available():
                return
            est, scaler_in, err_scaler = self._unwrap()
            if est is None or est.model is None or supported(est.model) is not None:
                return
            fwd = FusedHourglassForward(est.model, self.device)
            if scaler_in is not None:
                # MinMaxScaler: x' = x*scale_ + min_
                fwd.set_input_scaler(scaler_in.scale_, scaler_in.min_)
            self._err_scaler = err_scaler
            self._fused = fwd
            logger.info("Serving via fused HIP forward (model %s)", self.model_dir)
        except Exception as e:
            logger.warning("Fused serving path unavailable: %s", e)

    def _unwrap(self):
        """(estimator, input MinMaxScaler or None, anomaly scaler or None)."""
        from sklearn.pipeline import Pipeline
        from sklearn.preprocessing import MinMaxScaler

        from ..model.models import TorchAutoEncoder

        obj = self.model
        err_scaler = None
        if self.is_anomaly:
            err_scaler = getattr(obj, "scaler", None)
            obj = obj.base_estimator
        scaler = None
        if isinstance(obj, Pipeline):
            steps = [s for _, s in obj.steps]
            if len(steps) == 2 and isinstance(steps[0], MinMaxScaler):
                scaler, obj = steps[0], steps[1]
            elif len(steps) == 1:
                obj = steps[0]
            else:
                return None, None, None
        if not isinstance(obj, TorchAutoEncoder):
            return None, None, None
        return obj, scaler, err_scaler

    # ------------------------------------------------------------------
    def predict(self, X: pd.DataFrame) -> np.ndarray:
        if self._fused is not None:
            xt = torch.as_tensor(np.ascontiguousarray(X.values, dtype=np.float32))
            return self._fused.forward(xt).cpu().numpy()
        return self.model.predict(X.values)

    def anomaly(self, X: pd.DataFrame, y: Optional[pd.DataFrame], frequency=None) -> pd.DataFrame:
        if not self.is_anomaly:
            raise TypeError("Loaded model is not an anomaly detector")
        if y is None:
            y = X
        if self._fused is not None:
            # fused HIP forward for the expensive part; the detector's own
            # frame/threshold logic consumes the prediction unchanged
            xt = torch.as_tensor(np.ascontiguousarray(X.values, dtype=np.float32))
            out = self._fused.forward(xt).cpu().numpy()
            return self.model.anomaly_from_prediction(X, y, out, frequency)
        return self.model.anomaly(X, y, frequency)

    @property
    def tag_names(self):
        ds = self.metadata.get("dataset", {})
        return [t["name"] for t in ds.get("tag_list", [])]

    @property
    def model_offset(self) -> int:
        return int(self.metadata.get("model", {}).get("model-offset", 0) or 0)
