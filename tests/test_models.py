"""Model & factory tests (parity: tests/.../test_model.py — dim asserts,
fit/predict/score round-trip, sklearn get_params compliance, pickle
round-trip)."""
import pickle

import numpy as np
import pytest
from sklearn.base import clone

from gordo_amd.model.factories import (
    feedforward_hourglass,
    feedforward_model,
    feedforward_symmetric,
    lstm_hourglass,
    lstm_model,
)
from gordo_amd.model.factories.utils import hourglass_calc_dims
from gordo_amd.model.models import (
    TorchAutoEncoder,
    TorchLSTMAutoEncoder,
    TorchLSTMForecast,
)


def test_hourglass_calc_dims():
    assert hourglass_calc_dims(0.5, 2, 50) == [38, 25]
    assert hourglass_calc_dims(0.2, 3, 10) == [7, 5, 2]
    assert hourglass_calc_dims(0.5, 3, 3) == [3, 2, 2]
    assert hourglass_calc_dims(1.0, 3, 10) == [10, 10, 10]
    # never collapses below 1
    assert hourglass_calc_dims(0.0, 2, 4)[-1] == 1


def test_hourglass_dims_in_network():
    # Reference composition: hourglass == symmetric(hourglass_calc_dims), so
    # the bottleneck layer is duplicated (50-38-25-25-38-50).
    net = feedforward_hourglass(50, 50, encoding_layers=2, compression_factor=0.5)
    assert net.layer_widths == [50, 38, 25, 25, 38, 50]
    net = feedforward_hourglass(10, 10, encoding_layers=3, compression_factor=0.2)
    assert net.layer_widths == [10, 7, 5, 2, 2, 5, 7, 10]


def test_feedforward_model_shapes():
    net = feedforward_model(20, 20, encoding_dim=(16, 8), decoding_dim=(8, 16))
    assert net.layer_widths == [20, 16, 8, 8, 16, 20]


def test_feedforward_symmetric():
    net = feedforward_symmetric(30, 30, dims=(16, 8))
    assert net.layer_widths == [30, 16, 8, 8, 16, 30]


def test_lstm_factories():
    net = lstm_model(10, 10, lookback_window=12, encoding_dim=(8,), decoding_dim=(8,))
    assert net.lookback_window == 12
    net = lstm_hourglass(10, 10, encoding_layers=2, compression_factor=0.5)
    assert net.dims[0] > net.dims[1]


def test_autoencoder_fit_predict_score():
    rng = np.random.default_rng(0)
    X = rng.normal(size=(200, 10)).astype(np.float32)
    model = TorchAutoEncoder(kind="feedforward_hourglass", epochs=3, batch_size=64, seed=0)
    model.fit(X)
    out = model.predict(X)
    assert out.shape == X.shape
    s = model.score(X)
    assert isinstance(s, float)
    meta = model.get_metadata()
    assert "history" in meta and len(meta["history"]["loss"]) == 3


def test_autoencoder_sklearn_params_clone():
    model = TorchAutoEncoder(kind="feedforward_hourglass", epochs=2, compression_factor=0.3)
    params = model.get_params()
    assert params["kind"] == "feedforward_hourglass"
    assert params["compression_factor"] == 0.3
    cloned = clone(model)
    assert cloned.get_params() == params


def test_autoencoder_pickle_roundtrip():
    X = np.random.default_rng(1).normal(size=(100, 6)).astype(np.float32)
    model = TorchAutoEncoder(kind="feedforward_hourglass", epochs=1, seed=0)
    model.fit(X)
    blob = pickle.dumps(model)
    model2 = pickle.loads(blob)
    np.testing.assert_allclose(model.predict(X), model2.predict(X), rtol=1e-5, atol=1e-6)


def test_unknown_kind_raises():
    with pytest.raises(ValueError):
        TorchAutoEncoder(kind="not_a_real_kind")


def test_lstm_autoencoder_fit_predict():
    rng = np.random.default_rng(2)
    X = rng.normal(size=(80, 4)).astype(np.float32)
    model = TorchLSTMAutoEncoder(
        kind="lstm_hourglass", lookback_window=8, epochs=1, batch_size=16
    )
    model.fit(X)
    out = model.predict(X)
    assert out.shape == (80 - 8 + 1, 4)
    assert np.isfinite(out).all()


def test_lstm_forecast_fit_score():
    rng = np.random.default_rng(3)
    X = rng.normal(size=(60, 3)).astype(np.float32)
    model = TorchLSTMForecast(
        kind="lstm_symmetric", lookback_window=6, dims=(8,), epochs=1, batch_size=8
    )
    model.fit(X)
    s = model.score(X)
    assert isinstance(s, float)


def test_lstm_too_few_rows_raises():
    X = np.zeros((4, 3), dtype=np.float32)
    model = TorchLSTMAutoEncoder(kind="lstm_hourglass", lookback_window=8, epochs=1)
    with pytest.raises(ValueError):
        model.fit(X)


def test_wide_lstm_gemmloop_matches_module():
    """wide_forward (the GEMM-loop LSTM) must equal nn.LSTM numerically."""
    import torch

    from gordo_amd.model.torch_nets import LSTMNet
    from gordo_amd.ops.wide_lstm import wide_forward

    torch.manual_seed(0)
    net = LSTMNet(12, 12, [20, 8], [None, None], "linear", lookback_window=9)
    x = torch.randn(7, 9, 12)
    ref = net(x)
    out = wide_forward(net, x)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()


def test_wide_lstm_trainer_cpu():
    import numpy as np
    import torch

    from gordo_amd.model.torch_nets import LSTMNet
    from gordo_amd.ops.wide_lstm import WideLSTMTrainer
    from gordo_amd.runtime.trainer import TrainConfig

    torch.manual_seed(1)
    net = LSTMNet(6, 6, [10], [None], "linear", lookback_window=5)
    X = np.random.default_rng(0).normal(size=(200, 6)).astype(np.float32)
    cfg = TrainConfig(epochs=2, batch_size=64, device="cpu", seed=0)
    hist = WideLSTMTrainer(net, torch.device("cpu"), cfg).fit(X, X, lookback=5)
    assert hist.engine == "wide_lstm_gemmloop"
    assert len(hist.loss) == 2 and np.isfinite(hist.loss).all()


def test_forecast_predict_drops_out_of_range_windows():
    """A forecast model emits len(X) - lookback + 1 - lookahead rows: the
    final window (whose 1-step-ahead target is beyond X) is dropped so
    trailing-offset alignment downstream pairs prediction i with
    y[i + lookback - 1 + lookahead] (forecast misalignment)."""
    rng = np.random.default_rng(4)
    X = rng.normal(size=(40, 3)).astype(np.float32)
    lookback = 6
    model = TorchLSTMForecast(
        kind="lstm_symmetric", lookback_window=lookback, dims=(8,), epochs=1,
        batch_size=8,
    )
    model.fit(X)
    out = model.predict(X)
    assert out.shape == (len(X) - lookback + 1 - 1, 3)
    # AE (lookahead=0) keeps every window
    ae = TorchLSTMAutoEncoder(
        kind="lstm_symmetric", lookback_window=lookback, dims=(8,), epochs=1,
        batch_size=8,
    )
    ae.fit(X)
    assert ae.predict(X).shape == (len(X) - lookback + 1, 3)


def test_forecast_anomaly_alignment_oracle():
    """End-to-end alignment proof: an oracle module that perfectly predicts
    X[t+1] from the window ending at t must produce ~zero anomaly error
    through DiffBasedAnomalyDetector's trailing-offset alignment. A one-step
    misalignment (the r1 bug) would make the error the step size (~1.0)."""
    import torch

    from gordo_amd.model.anomaly.diff import DiffBasedAnomalyDetector

    lookback = 5
    n, f = 30, 2
    # strictly increasing ramp: X[t] = t (per feature), so X[t+1] = X[t] + 1
    X = np.tile(np.arange(n, dtype=np.float32)[:, None], (1, f))

    class Oracle(torch.nn.Module):
        lookback_window = lookback

        def forward(self, w):  # w: [B, T, F] -> predict next row after window
            return w[:, -1, :] + 1.0

    est = TorchLSTMForecast(kind="lstm_symmetric", lookback_window=lookback)
    est.model = Oracle()
    est._n_features = f
    est._n_features_out = f

    out = est.predict(X)
    assert out.shape == (n - lookback, f)
    targets = X[lookback - 1 + 1 :]
    np.testing.assert_allclose(out, targets, atol=1e-6)
    assert est.score(X) == pytest.approx(1.0)

    class IdentityScaler:
        def fit(self, y):
            return self

        def transform(self, y):
            return np.asarray(y)

    det = DiffBasedAnomalyDetector(base_estimator=est, scaler=IdentityScaler())
    frame = det.anomaly(X, X)
    err = frame["tag-anomaly-scores"].values
    np.testing.assert_allclose(err, 0.0, atol=1e-5)
