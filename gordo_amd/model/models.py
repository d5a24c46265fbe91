"""sklearn-compatible torch estimators.

Parity: gordo_components/model/models.py — ``KerasBaseEstimator`` and friends,
re-based on PyTorch-ROCm:

- ``TorchBaseEstimator(kind, **kwargs)``: resolves ``kind`` through the factory
  registry, builds the network lazily from input dims at fit time, trains via
  gordo_amd.runtime.Trainer (fused HIP kernels on MI355X, eager torch on CPU),
  and round-trips through pickle with weights embedded as bytes (the
  reference serializes Keras weights as HDF5 bytes inside __getstate__).
- ``TorchAutoEncoder``: reconstruction model, score = explained variance.
- ``TorchLSTMAutoEncoder`` / ``TorchLSTMForecast``: 3-D rolling-window models
  over ``lookback_window`` timesteps (lookahead 0 / 1).
"""
from __future__ import annotations

import io
import logging
import threading
from typing import Any, Dict, Optional

import numpy as np
import pandas as pd
import torch
from sklearn.base import BaseEstimator, TransformerMixin
from sklearn.metrics import explained_variance_score

from .base import GordoBase
from .register import lookup_factory
from ..runtime.trainer import TrainConfig, Trainer, resolve_device

logger = logging.getLogger(__name__)


class TorchBaseEstimator(BaseEstimator, GordoBase):
    """Base for all torch-backed gordo models.

    ``kind`` names a registered factory; every other kwarg is forwarded to the
    factory (architecture) and/or TrainConfig (fit params) — the same
    "one kwargs bag" convention the reference's KerasBaseEstimator uses.
    """

    lookahead: int = 0
    supports_lookback: bool = False

    def __init__(self, kind: str, **kwargs) -> None:
        self.kind = kind
        self.kwargs = kwargs
        self.model: Optional[torch.nn.Module] = None
        self.history: Optional[Dict[str, Any]] = None
        self._validate_kind()

    # -- sklearn plumbing -------------------------------------------------
    def _validate_kind(self):
        lookup_factory(type(self).__name__, self.kind)

    def get_params(self, deep=False) -> Dict[str, Any]:
        params = {"kind": self.kind}
        params.update(self.kwargs)
        return params

    def set_params(self, **params):
        self.kind = params.pop("kind", self.kind)
        self.kwargs.update(params)
        return self

    @property
    def sk_params(self) -> Dict[str, Any]:
        return self.kwargs

    # -- persistence -------------------------------------------------------
    # When _externalize_weights is set (serializer.dump's externalize()
    # context), pickles carry only a marker + dims; the tensors live in the
    # artifact dir's weights.safetensors and serializer.load() mmaps them
    # back in — the zero-pickle fast load path. Plain pickle/dumps stays
    # self-contained (weights embedded as bytes, the reference's Keras-HDF5-
    # inside-pickle idiom). Thread-local so a concurrent dumps() (e.g. the
    # server's /download-model) can never observe another thread's dump().
    _externalize_tls = threading.local()

    def __getstate__(self):
        state = self.__dict__.copy()
        if self.model is not None:
            if getattr(TorchBaseEstimator._externalize_tls, "on", False):
                state["model"] = {
                    "__external_weights__": True,
                    "n_features": getattr(self, "_n_features", None),
                    "n_features_out": getattr(self, "_n_features_out", None),
                }
                return state
            buf = io.BytesIO()
            torch.save(
                {
                    "state_dict": {
                        k: v.cpu() for k, v in self.model.state_dict().items()
                    },
                    "n_features": getattr(self, "_n_features", None),
                    "n_features_out": getattr(self, "_n_features_out", None),
                },
                buf,
            )
            state["model"] = buf.getvalue()
        return state

    def __setstate__(self, state):
        model_bytes = state.get("model")
        self.__dict__ = state
        if isinstance(model_bytes, dict) and model_bytes.get("__external_weights__"):
            self._n_features = model_bytes["n_features"]
            self._n_features_out = model_bytes["n_features_out"]
            self.model = None
            self._needs_external_weights = True
        elif isinstance(model_bytes, (bytes, bytearray)):
            payload = torch.load(io.BytesIO(model_bytes), map_location="cpu", weights_only=True)
            self._n_features = payload["n_features"]
            self._n_features_out = payload["n_features_out"]
            self.model = self._build(self._n_features, self._n_features_out)
            self.model.load_state_dict(payload["state_dict"])
            self.model.eval()

    def __sklearn_is_fitted__(self) -> bool:
        return self.model is not None

    def restore_weights(self, state_dict: Dict[str, torch.Tensor]) -> None:
        """Rebuild the module and load externally-stored tensors (the
        weights.safetensors half of an externalized artifact)."""
        self.model = self._build(self._n_features, self._n_features_out)
        self.model.load_state_dict(state_dict)
        self.model.eval()
        self.__dict__.pop("_needs_external_weights", None)

    # -- build / fit / predict ----------------------------------------------
    def _factory_kwargs(self) -> Dict[str, Any]:
        drop = set(TrainConfig.__dataclass_fields__)
        return {k: v for k, v in self.kwargs.items() if k not in drop}

    def _build(self, n_features: int, n_features_out: int) -> torch.nn.Module:
        factory = lookup_factory(type(self).__name__, self.kind)
        return factory(n_features, n_features_out, **self._factory_kwargs())

    @staticmethod
    def _as_array(X) -> np.ndarray:
        if isinstance(X, pd.DataFrame):
            X = X.values
        arr = np.asarray(X, dtype=np.float32)
        if arr.ndim == 1:
            arr = arr.reshape(-1, 1)
        return arr

    @property
    def lookback_window(self) -> int:
        return int(self.kwargs.get("lookback_window", 144)) if self.supports_lookback else 1

    def fit(self, X, y=None, **fit_args):
        X = self._as_array(X)
        y = X if y is None else self._as_array(y)
        if len(y) != len(X):
            raise ValueError(f"X and y row counts differ: {len(X)} vs {len(y)}")

        self._n_features = X.shape[1]
        self._n_features_out = y.shape[1]
        # sklearn's standard fitted marker: without a trailing-underscore
        # attribute, check_is_fitted() deems the estimator unfitted even
        # after fit() (a hard error inside Pipeline.predict from sklearn 1.8)
        self.n_features_in_ = X.shape[1]
        if self.model is None:
            # seed BEFORE the build so weight init is deterministic too (the
            # trainer re-seeds before shuffling; without this the first fit
            # in a process inherits whatever global RNG state preceded it)
            seed = {**self.kwargs, **fit_args}.get("seed")
            if seed is not None:
                torch.manual_seed(int(seed))
            self.model = self._build(self._n_features, self._n_features_out)

        cfg = TrainConfig.from_kwargs({**self.kwargs, **fit_args})
        trainer = Trainer(cfg)
        lookback = self.lookback_window if self.supports_lookback else 0
        history = trainer.fit(
            self.model, X, y, lookback=lookback, lookahead=self.lookahead
        )
        self.history = history.to_dict()
        return self

    def _predict_array(self, X: np.ndarray, batch_size: int = 65536) -> np.ndarray:
        if self.model is None:
            if getattr(self, "_needs_external_weights", False):
                raise ValueError(
                    "Estimator was unpickled from an externalized artifact; "
                    "load it via serializer.load(dir) so weights.safetensors "
                    "is restored"
                )
            raise ValueError("Model has not been fit yet")
        device = resolve_device(self.kwargs.get("device", "auto"))
        model = self.model.to(device)
        model.eval()
        outs = []
        with torch.no_grad():
            for s in range(0, len(X), batch_size):
                xb = torch.as_tensor(X[s : s + batch_size], device=device)
                outs.append(model(xb).float().cpu().numpy())
        return np.concatenate(outs, axis=0) if outs else np.empty((0, self._n_features_out))

    def get_metadata(self) -> Dict[str, Any]:
        meta: Dict[str, Any] = {}
        if self.history is not None:
            meta["history"] = self.history
        return meta

    def score(self, X, y=None, sample_weight=None) -> float:
        X = self._as_array(X)
        y = X if y is None else self._as_array(y)
        out = self.predict(X)
        offset = len(y) - len(out)
        return explained_variance_score(y[offset:], out)


class TorchAutoEncoder(TorchBaseEstimator, TransformerMixin):
    """Dense autoencoder (parity: KerasAutoEncoder)."""

    def predict(self, X) -> np.ndarray:
        X = self._as_array(X)
        return self._predict_array(X)

    def transform(self, X) -> np.ndarray:
        """TransformerMixin contract (FeatureUnion/pipeline middle steps):
        the autoencoder's transform IS its reconstruction."""
        return self.predict(X)


class TorchLSTMBaseEstimator(TorchBaseEstimator, TransformerMixin):
    supports_lookback = True

    def _window_predict(self, X: np.ndarray, batch_size: int = 8192) -> np.ndarray:
        if self.model is None:
            raise ValueError("Model has not been fit yet")
        lookback = self.lookback_window
        if len(X) < lookback:
            raise ValueError(
                f"Need at least lookback_window={lookback} rows, got {len(X)}"
            )
        device = resolve_device(self.kwargs.get("device", "auto"))
        use_wide = False
        if device.type == "cuda":
            from ..ops import is_available
            from ..ops.lstm_engine import FusedLSTMForward, lstm_supported

            if is_available() and lstm_supported(self.model) is None:
                fwd = FusedLSTMForward(self.model, device)
                return fwd.predict(X, lookahead=self.lookahead)
            use_wide = True  # GEMM-loop forward instead of nn.LSTM eager
        model = self.model.to(device)
        model.eval()
        Xt = torch.as_tensor(X, device=device)
        # Drop trailing windows whose target row (window end + lookahead) is
        # beyond X: output row i always pairs with y[i + lookback - 1 +
        # lookahead] under the trailing-offset alignment used by diff.py /
        # metric_wrapper / make_base_dataframe. Matches the reference, whose
        # forecast predict emits len(X) - lookback rows.
        n_windows = len(X) - lookback + 1 - self.lookahead
        offsets = torch.arange(lookback, device=device)
        outs = []
        with torch.no_grad():
            for s in range(0, n_windows, batch_size):
                idx = torch.arange(s, min(s + batch_size, n_windows), device=device)
                w = Xt[idx.unsqueeze(1) + offsets.unsqueeze(0)]
                if use_wide:
                    from ..ops.wide_lstm import wide_forward

                    out = wide_forward(model, w)
                else:
                    out = model(w)
                outs.append(out.float().cpu().numpy())
        return np.concatenate(outs, axis=0)

    def predict(self, X) -> np.ndarray:
        X = self._as_array(X)
        return self._window_predict(X)

    def transform(self, X) -> np.ndarray:
        """TransformerMixin contract: windowed reconstruction/forecast."""
        return self.predict(X)


class TorchLSTMAutoEncoder(TorchLSTMBaseEstimator):
    """Reconstructs X[t] from the window X[t-lookback+1 .. t]
    (parity: KerasLSTMAutoEncoder). predict() output has
    len(X) - lookback + 1 rows, aligned to the window ends."""

    lookahead = 0


class TorchLSTMForecast(TorchLSTMBaseEstimator):
    """Forecasts X[t+1] from the window ending at t
    (parity: KerasLSTMForecast, lookahead=1). predict() emits
    len(X) - lookback_window rows — prediction i targets
    X[i + lookback - 1 + lookahead], so trailing-offset alignment in
    diff.py / metric_wrapper pairs every prediction with its true target."""

    lookahead = 1

    def score(self, X, y=None, sample_weight=None) -> float:
        X = self._as_array(X)
        y = X if y is None else self._as_array(y)
        out = self.predict(X)
        # prediction i targets row i + lookback - 1 + lookahead; predict()
        # already drops the windows whose target is beyond X, so lengths match
        targets = y[self.lookback_window - 1 + self.lookahead :]
        assert len(targets) == len(out), (len(targets), len(out))
        return explained_variance_score(targets, out)
