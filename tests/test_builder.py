"""Builder tests (parity: tests/.../test_build_model.py — real tiny builds
into tmpdir, metadata schema, cache hit on second build)."""
import numpy as np
import pandas as pd
import pytest

from gordo_amd.builder import ModelBuilder
from gordo_amd.machine import Machine
from gordo_amd.serializer import load, load_metadata


@pytest.fixture
def machine(machine_config):
    return Machine.from_config(machine_config)


def test_build_produces_model_and_metadata(machine, tmp_path):
    out = tmp_path / "model"
    model, metadata = ModelBuilder(machine).build(output_dir=str(out))
    assert (out / "model.pkl").exists()
    assert (out / "metadata.json").exists()

    assert metadata["name"] == "test-machine"
    assert "cross-validation" in metadata["model"]
    scores = metadata["model"]["cross-validation"]["scores"]
    assert "explained_variance_score" in scores
    assert "mean" in scores["explained_variance_score"]
    assert metadata["model"]["cross-validation"]["cv-duration-sec"] > 0
    assert metadata["dataset"]["data-query-duration-sec"] > 0
    assert "thresholds" in metadata["model"]["cross-validation"]

    loaded = load(str(out))
    X = np.random.default_rng(0).normal(size=(20, 10)).astype(np.float32)
    df = pd.DataFrame(X, columns=[f"tag-{i}" for i in range(10)])
    out_frame = loaded.anomaly(df, df)
    assert ("total-anomaly-score", "") in out_frame.columns


def test_build_cache_hit(machine, tmp_path):
    reg = tmp_path / "registry"
    out1 = tmp_path / "m1"
    out2 = tmp_path / "m2"
    builder = ModelBuilder(machine)
    _, meta1 = builder.build(output_dir=str(out1), model_register_dir=str(reg))
    # second build with same config: cache hit (same artifact copied)
    model2, meta2 = builder.build(output_dir=str(out2), model_register_dir=str(reg))
    assert (out2 / "model.pkl").exists()
    assert meta2["name"] == meta1["name"]
    # cached metadata reused verbatim (no retraining)
    assert (
        meta2["model"]["model-training-duration-sec"]
        == meta1["model"]["model-training-duration-sec"]
    )


def test_cache_key_stable_and_sensitive(machine, machine_config):
    key1 = ModelBuilder(machine).cache_key
    key2 = ModelBuilder(Machine.from_config(machine_config)).cache_key
    assert key1 == key2
    machine_config2 = dict(machine_config)
    machine_config2["dataset"] = dict(machine_config["dataset"], resolution="1h")
    key3 = ModelBuilder(Machine.from_config(machine_config2)).cache_key
    assert key3 != key1


def test_cv_mode_cross_val_only(machine_config, tmp_path):
    machine_config = dict(machine_config)
    machine_config["evaluation"] = {"cv_mode": "cross_val_only"}
    machine = Machine.from_config(machine_config)
    model, metadata = ModelBuilder(machine).build()
    assert "cross-validation" in metadata["model"]
    assert "model-training-duration-sec" not in metadata["model"]


def test_model_offset_for_lstm(machine_config):
    machine_config = dict(machine_config)
    machine_config["model"] = {
        "gordo_amd.model.models.TorchLSTMAutoEncoder": {
            "kind": "lstm_hourglass",
            "lookback_window": 8,
            "epochs": 1,
        }
    }
    machine = Machine.from_config(machine_config)
    model, metadata = ModelBuilder(machine).build()
    assert metadata["model"]["model-offset"] == 7


def test_cv_per_tag_scores(tmp_path):
    """CV metadata carries per-tag scores next to the aggregate (parity with
    the reference's metric_wrapper-expanded scorer dict)."""
    from gordo_amd.builder import ModelBuilder
    from gordo_amd.machine import Machine

    config = {
        "name": "per-tag",
        "dataset": {
            "type": "RandomDataset",
            "train_start_date": "2020-01-01T00:00:00Z",
            "train_end_date": "2020-01-20T00:00:00Z",
            "tag_list": [f"tag-{i}" for i in range(3)],
        },
        "model": {
            "gordo_amd.model.models.TorchAutoEncoder": {
                "kind": "feedforward_hourglass", "epochs": 1, "batch_size": 64
            }
        },
        "evaluation": {"metrics": ["explained_variance_score"]},
    }
    _, meta = ModelBuilder(Machine.from_config(config)).build(output_dir=str(tmp_path / "m"))
    scores = meta["model"]["cross-validation"]["scores"]
    assert "explained_variance_score" in scores
    for i in range(3):
        assert f"explained_variance_score-tag-{i}" in scores


# ---------------------------------------------------------------------------
# Torch-vectorized CV scoring (builder/cv.py, K9) vs sklearn ground truth
# ---------------------------------------------------------------------------

def test_fold_scores_match_sklearn():
    from sklearn.metrics import (
        explained_variance_score,
        mean_absolute_error,
        mean_squared_error,
        r2_score,
    )
    from sklearn.preprocessing import RobustScaler

    from gordo_amd.builder.cv import fold_scores

    rng = np.random.default_rng(0)
    y = rng.normal(size=(300, 7))
    pred = y + rng.normal(0, 0.3, size=(300, 7))
    scaler = RobustScaler().fit(y)
    names = [f"t{i}" for i in range(7)]
    metrics = [
        "explained_variance_score",
        "r2_score",
        "mean_squared_error",
        "mean_absolute_error",
    ]
    got = fold_scores(y, pred, metrics, scaler=scaler, tag_names=names)
    ys, ps = scaler.transform(y), scaler.transform(pred)
    for fname, func in [
        ("explained_variance_score", explained_variance_score),
        ("r2_score", r2_score),
        ("mean_squared_error", mean_squared_error),
        ("mean_absolute_error", mean_absolute_error),
    ]:
        assert got[fname] == pytest.approx(func(ys, ps), rel=1e-10)
        for i, t in enumerate(names):
            assert got[f"{fname}-{t}"] == pytest.approx(
                func(ys[:, i], ps[:, i]), rel=1e-10
            ), (fname, t)


def test_fold_scores_zero_variance_conventions():
    from sklearn.metrics import explained_variance_score, r2_score

    from gordo_amd.builder.cv import fold_scores

    # constant y_true column: sklearn EV/R2 -> 0.0 when prediction differs,
    # 1.0 when prediction is exact
    y = np.ones((50, 1))
    exact = np.ones((50, 1))
    off = np.full((50, 1), 2.0)
    m = ["explained_variance_score", "r2_score"]
    assert fold_scores(y, exact, m, per_tag=False)["r2_score"] == 1.0
    assert fold_scores(y, off, m, per_tag=False)["r2_score"] == r2_score(y, off)
    assert (
        fold_scores(y, off, m, per_tag=False)["explained_variance_score"]
        == explained_variance_score(y, off)
    )


def test_fold_scores_trailing_offset_alignment():
    from gordo_amd.builder.cv import fold_scores

    y = np.arange(20, dtype=float)[:, None]
    pred = y[5:]  # lookback model: fewer prediction rows, aligned to the end
    s = fold_scores(y, pred, ["mean_absolute_error"], per_tag=False)
    assert s["mean_absolute_error"] == 0.0


def test_run_time_series_cv_shapes_and_estimators():
    from sklearn.model_selection import TimeSeriesSplit

    from gordo_amd.builder.cv import run_time_series_cv
    from gordo_amd.model.models import TorchAutoEncoder

    rng = np.random.default_rng(1)
    X = rng.normal(size=(200, 5)).astype(np.float32)
    model = TorchAutoEncoder(kind="feedforward_hourglass", epochs=1, batch_size=64, seed=0)
    out = run_time_series_cv(
        model, X, X, cv=TimeSeriesSplit(n_splits=3),
        metric_names=["explained_variance_score", "mean_squared_error"],
        tag_names=[f"t{i}" for i in range(5)],
        return_estimator=True, return_predictions=True,
    )
    assert len(out["test_explained_variance_score"]) == 3
    assert "test_mean_squared_error-t4" in out
    assert len(out["estimator"]) == 3
    assert len(out["predictions"]) == 3
    idx, pred = out["predictions"][0]
    assert pred.shape == (len(idx), 5)
    # the original model must be untouched (clones are fit)
    assert model.model is None
