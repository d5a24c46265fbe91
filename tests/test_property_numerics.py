"""Property-based numerics tests (hypothesis): the torch-vectorized CV
scorer vs sklearn ground truth over randomized shapes/values, and
rolling-window invariants. Complements the fixed-seed unit tests."""
import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

from gordo_amd.builder.cv import fold_scores
from gordo_amd.model.utils import rolling_windows


@settings(max_examples=30, deadline=None)
@given(
    n=st.integers(min_value=3, max_value=200),
    f=st.integers(min_value=1, max_value=12),
    seed=st.integers(min_value=0, max_value=2**31 - 1),
    noise=st.floats(min_value=0.0, max_value=5.0),
)
def test_fold_scores_match_sklearn_property(n, f, seed, noise):
    from sklearn.metrics import (
        explained_variance_score,
        mean_absolute_error,
        mean_squared_error,
        r2_score,
    )

    rng = np.random.default_rng(seed)
    y = rng.normal(size=(n, f))
    pred = y + noise * rng.normal(size=(n, f))
    metrics = [
        "explained_variance_score",
        "r2_score",
        "mean_squared_error",
        "mean_absolute_error",
    ]
    got = fold_scores(y, pred, metrics, per_tag=True, tag_names=[str(i) for i in range(f)])
    for name, func in [
        ("explained_variance_score", explained_variance_score),
        ("r2_score", r2_score),
        ("mean_squared_error", mean_squared_error),
        ("mean_absolute_error", mean_absolute_error),
    ]:
        ref = func(y, pred)
        assert got[name] == pytest.approx(ref, rel=1e-9, abs=1e-12), name
        for i in range(f):
            assert got[f"{name}-{i}"] == pytest.approx(
                func(y[:, i], pred[:, i]), rel=1e-9, abs=1e-12
            )


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(min_value=2, max_value=300),
    f=st.integers(min_value=1, max_value=8),
    lookback=st.integers(min_value=1, max_value=48),
    lookahead=st.integers(min_value=0, max_value=3),
)
def test_rolling_windows_invariants(n, f, lookback, lookahead):
    X = np.arange(n * f, dtype=np.float32).reshape(n, f)
    if n - lookback + 1 - lookahead <= 0:
        with pytest.raises(ValueError):
            rolling_windows(X, lookback, lookahead)
        return
    w, t = rolling_windows(X, lookback, lookahead)
    n_windows = n - lookback + 1 - lookahead
    assert w.shape == (n_windows, lookback, f)
    assert t.shape[0] >= n_windows
    # window i is exactly X[i : i+lookback]; target i = X[i+lookback-1+lookahead]
    for i in (0, n_windows // 2, n_windows - 1):
        np.testing.assert_array_equal(w[i], X[i : i + lookback])
        np.testing.assert_array_equal(t[i], X[i + lookback - 1 + lookahead])
    # zero-copy: the windows view shares memory with X
    assert np.shares_memory(w, X)
