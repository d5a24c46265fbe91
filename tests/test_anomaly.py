"""CPU behavior tests for DiffBasedAnomalyDetector: frame structure,
smoothed scores, CV-derived thresholds/confidences, require_thresholds."""
import numpy as np
import pandas as pd
import pytest

from gordo_amd.model.anomaly.diff import DiffBasedAnomalyDetector
from gordo_amd.model.models import TorchAutoEncoder


def _fit_detector(**kwargs):
    rng = np.random.default_rng(0)
    Z = rng.normal(size=(600, 4)).astype(np.float32)
    P = rng.normal(size=(4, 8)).astype(np.float32)
    X = np.tanh(Z @ P)
    det = DiffBasedAnomalyDetector(
        base_estimator=TorchAutoEncoder(
            kind="feedforward_hourglass", encoding_layers=1, epochs=2, batch_size=128
        ),
        **kwargs,
    )
    det.fit(X)
    det.scaler.fit(X)
    return det, X


def test_anomaly_frame_structure():
    det, X = _fit_detector()
    df = det.anomaly(pd.DataFrame(X, columns=[f"t{i}" for i in range(8)]),
                     pd.DataFrame(X, columns=[f"t{i}" for i in range(8)]))
    tops = set(df.columns.get_level_values(0))
    assert {"model-input", "model-output", "tag-anomaly-scores",
            "total-anomaly-score"} <= tops
    tot = df[("total-anomaly-score", "")].values
    err = df["tag-anomaly-scores"].values
    np.testing.assert_allclose(tot, np.linalg.norm(err, axis=1), rtol=1e-5)


def test_smoothed_score_columns():
    det, X = _fit_detector(window=5)
    df = det.anomaly(pd.DataFrame(X, columns=[f"t{i}" for i in range(8)]),
                     pd.DataFrame(X, columns=[f"t{i}" for i in range(8)]))
    tops = set(df.columns.get_level_values(0))
    assert "smooth-tag-anomaly-scores" in tops
    assert "smooth-total-anomaly-score" in tops
    # the smoothed score is the rolling mean of the raw per-tag scores
    raw = df["tag-anomaly-scores"]
    sm = df["smooth-tag-anomaly-scores"]
    expected = raw.rolling(5, min_periods=1).mean()
    np.testing.assert_allclose(sm.values, expected.values, rtol=1e-5, atol=1e-7)


def test_cv_thresholds_and_confidences():
    det, X = _fit_detector()
    Xf = pd.DataFrame(
        X, columns=[f"t{i}" for i in range(8)],
        index=pd.date_range("2021-01-01", periods=len(X), freq="10min"),
    )
    det.cross_validate(X=Xf, y=Xf)
    assert det.feature_thresholds_ is not None and len(det.feature_thresholds_) == 8
    assert det.aggregate_threshold_ is not None and det.aggregate_threshold_ > 0
    meta = det.get_metadata()
    assert "feature-thresholds" in meta and "aggregate-threshold" in meta
    df = det.anomaly(Xf, Xf)
    tops = set(df.columns.get_level_values(0))
    assert "anomaly-confidence" in tops and "total-anomaly-confidence" in tops
    conf = df[("total-anomaly-confidence", "")].values
    tot = df[("total-anomaly-score", "")].values
    np.testing.assert_allclose(conf, tot / det.aggregate_threshold_, rtol=1e-6)


def test_require_thresholds_enforced():
    det, X = _fit_detector(require_thresholds=True)
    with pytest.raises(AttributeError, match="thresholds"):
        det.anomaly(pd.DataFrame(X, columns=[f"t{i}" for i in range(8)]),
                    pd.DataFrame(X, columns=[f"t{i}" for i in range(8)]))


def test_metric_wrapper_scaling_and_per_tag():
    """metric_wrapper transforms through the scaler before scoring and the
    col argument restricts scoring to one tag."""
    from sklearn.metrics import mean_squared_error
    from sklearn.preprocessing import StandardScaler

    from gordo_amd.model.utils import metric_wrapper

    rng = np.random.default_rng(1)
    y_true = rng.normal(0, [1.0, 100.0], size=(200, 2))
    y_pred = y_true + rng.normal(0, [0.1, 10.0], size=(200, 2))
    scaler = StandardScaler().fit(y_true)

    plain = mean_squared_error(y_true, y_pred)
    scaled = metric_wrapper(mean_squared_error, scaler)(y_true, y_pred)
    expected = mean_squared_error(scaler.transform(y_true), scaler.transform(y_pred))
    assert abs(scaled - expected) < 1e-9
    assert abs(scaled - plain) > 1.0  # the big-magnitude tag dominated plain MSE

    per_tag0 = metric_wrapper(mean_squared_error, scaler, col=0)(y_true, y_pred)
    per_tag1 = metric_wrapper(mean_squared_error, scaler, col=1)(y_true, y_pred)
    s_true, s_pred = scaler.transform(y_true), scaler.transform(y_pred)
    assert abs(per_tag0 - mean_squared_error(s_true[:, 0], s_pred[:, 0])) < 1e-9
    assert abs(per_tag1 - mean_squared_error(s_true[:, 1], s_pred[:, 1])) < 1e-9


def test_metric_wrapper_lookback_alignment():
    """Lookback models emit fewer rows than y: the wrapper aligns to window
    ends before scoring."""
    from sklearn.metrics import mean_absolute_error

    from gordo_amd.model.utils import metric_wrapper

    y_true = np.arange(20, dtype=float).reshape(10, 2)
    y_pred = y_true[3:] + 1.0  # 7 rows, as if lookback=4
    got = metric_wrapper(mean_absolute_error)(y_true, y_pred)
    assert abs(got - 1.0) < 1e-12


def test_transformer_funcs_in_yaml_pipeline():
    """InfImputer / multiply are YAML-referenceable and behave (parity:
    gordo_components/model/transformer_funcs)."""
    from gordo_amd.model.transformer_funcs.general import InfImputer, multiply
    from gordo_amd.serializer import from_definition

    X = np.array([[1.0, np.inf], [-np.inf, 2.0], [3.0, 4.0]])
    imp = InfImputer().fit(X)
    out = imp.transform(X)
    assert np.isfinite(out).all()
    assert out[0, 1] == 4.0 + 2.0 and out[1, 0] == 1.0 - 2.0

    imp2 = InfImputer(inf_fill_value=99.0, neg_inf_fill_value=-99.0).fit(X)
    out2 = imp2.transform(X)
    assert out2[0, 1] == 99.0 and out2[1, 0] == -99.0

    assert (multiply(np.ones((2, 2)), factor=3.0) == 3.0).all()

    # resolvable from a pipeline definition, like the reference YAMLs
    pipe = from_definition({
        "sklearn.pipeline.Pipeline": {
            "steps": [
                "gordo_amd.model.transformer_funcs.general.InfImputer",
                "sklearn.preprocessing.MinMaxScaler",
            ]
        }
    })
    got = pipe.fit_transform(X)
    assert np.isfinite(got).all() and got.min() >= 0 and got.max() <= 1


def test_rolling_windows_lookahead_semantics():
    """rolling_windows: windows[i] = X[i:i+lookback], target[i] =
    X[i+lookback-1+lookahead] (forecast = 1-step-ahead)."""
    from gordo_amd.model.utils import rolling_windows

    X = np.arange(10, dtype=np.float32)[:, None]
    w, t = rolling_windows(X, lookback=3, lookahead=0)
    assert w.shape == (8, 3, 1) and t.shape == (8, 1)
    assert (w[0].ravel() == [0, 1, 2]).all() and t[0, 0] == 2  # AE: last row
    w1, t1 = rolling_windows(X, lookback=3, lookahead=1)
    assert w1.shape == (7, 3, 1)
    assert (w1[0].ravel() == [0, 1, 2]).all() and t1[0, 0] == 3  # forecast
    with pytest.raises(Exception):
        rolling_windows(X[:2], lookback=3, lookahead=0)
