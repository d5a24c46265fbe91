"""GPU serving-path tests: the fused HIP anomaly route must match the eager
detector output on the same artifact."""
import numpy as np
import pandas as pd
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)


@pytest.fixture(scope="module")
def runtime_pair(tmp_path_factory):
    from gordo_amd.builder import ModelBuilder
    from gordo_amd.machine import Machine
    from gordo_amd.server.model_io import ModelRuntime

    config = {
        "name": "serving-parity",
        "dataset": {
            "type": "RandomDataset",
            "train_start_date": "2020-01-01T00:00:00Z",
            "train_end_date": "2020-02-01T00:00:00Z",
            "tag_list": [f"tag-{i}" for i in range(12)],
        },
        "model": {
            "gordo_amd.model.anomaly.diff.DiffBasedAnomalyDetector": {
                "base_estimator": {
                    "sklearn.pipeline.Pipeline": {
                        "steps": [
                            "sklearn.preprocessing.MinMaxScaler",
                            {
                                "gordo_amd.model.models.TorchAutoEncoder": {
                                    "kind": "feedforward_hourglass",
                                    "encoding_layers": 2,
                                    "epochs": 3,
                                    "batch_size": 512,
                                }
                            },
                        ]
                    }
                }
            }
        },
    }
    out = tmp_path_factory.mktemp("model") / "serving-parity"
    ModelBuilder(Machine.from_config(config)).build(output_dir=str(out))
    fused = ModelRuntime(str(out))
    eager = ModelRuntime(str(out))
    eager._fused = None
    return fused, eager


def test_fused_anomaly_matches_eager(runtime_pair):
    fused, eager = runtime_pair
    assert fused._fused is not None, "fused serving path not active on GPU"
    rng = np.random.default_rng(0)
    X = pd.DataFrame(
        rng.uniform(0, 1, size=(300, 12)).astype(np.float32),
        columns=[f"tag-{i}" for i in range(12)],
        index=pd.date_range("2021-01-01", periods=300, freq="10min"),
    )
    df_f = fused.anomaly(X, None)
    df_e = eager.anomaly(X, None)
    assert list(df_f.columns) == list(df_e.columns)
    tot_f = df_f[("total-anomaly-score", "")].values
    tot_e = df_e[("total-anomaly-score", "")].values
    # fused forward is bf16; eager is fp32
    np.testing.assert_allclose(tot_f, tot_e, rtol=5e-2, atol=5e-2)
    out_f = df_f["model-output"].values
    out_e = df_e["model-output"].values
    np.testing.assert_allclose(out_f, out_e, rtol=5e-2, atol=3e-2)


def test_fused_predict_matches_eager(runtime_pair):
    fused, eager = runtime_pair
    rng = np.random.default_rng(1)
    X = pd.DataFrame(
        rng.uniform(0, 1, size=(200, 12)).astype(np.float32),
        columns=[f"tag-{i}" for i in range(12)],
    )
    np.testing.assert_allclose(
        fused.predict(X), eager.predict(X), rtol=5e-2, atol=3e-2
    )


def test_x256_model_serving_fused(tmp_path):
    """200-tag machine (W=256 tier): builder -> artifact -> fused serving,
    with anomaly parity against the eager detector."""
    from gordo_amd.builder import ModelBuilder
    from gordo_amd.machine import Machine
    from gordo_amd.server.model_io import ModelRuntime

    config = {
        "name": "x256-serving",
        "dataset": {
            "type": "RandomDataset",
            "train_start_date": "2020-01-01T00:00:00Z",
            "train_end_date": "2020-01-10T00:00:00Z",
            "tag_list": [f"tag-{i}" for i in range(200)],
        },
        "model": {
            "gordo_amd.model.anomaly.diff.DiffBasedAnomalyDetector": {
                "base_estimator": {
                    "sklearn.pipeline.Pipeline": {
                        "steps": [
                            "sklearn.preprocessing.MinMaxScaler",
                            {
                                "gordo_amd.model.models.TorchAutoEncoder": {
                                    "kind": "feedforward_hourglass",
                                    "encoding_layers": 2,
                                    "epochs": 2,
                                    "batch_size": 512,
                                }
                            },
                        ]
                    }
                }
            }
        },
        "evaluation": {"cv_mode": "no_cv"},
    }
    out = tmp_path / "x256-serving"
    ModelBuilder(Machine.from_config(config)).build(output_dir=str(out))
    fused = ModelRuntime(str(out))
    assert fused._fused is not None, "W=256 fused serving path not active"
    assert fused._fused.p.wmax == 256
    eager = ModelRuntime(str(out))
    eager._fused = None
    rng = np.random.default_rng(0)
    X = pd.DataFrame(
        rng.uniform(0, 1, size=(200, 200)).astype(np.float32),
        columns=[f"tag-{i}" for i in range(200)],
    )
    df_f = fused.anomaly(X, None)
    df_e = eager.anomaly(X, None)
    tot_f = df_f[("total-anomaly-score", "")].values
    tot_e = df_e[("total-anomaly-score", "")].values
    np.testing.assert_allclose(tot_f, tot_e, rtol=6e-2, atol=6e-2)


def test_wide_model_serving_fused(tmp_path):
    """100-tag machine: builder -> artifact -> fused wide serving."""
    from gordo_amd.builder import ModelBuilder
    from gordo_amd.machine import Machine
    from gordo_amd.server.model_io import ModelRuntime

    config = {
        "name": "wide-serving",
        "dataset": {
            "type": "RandomDataset",
            "train_start_date": "2020-01-01T00:00:00Z",
            "train_end_date": "2020-01-15T00:00:00Z",
            "tag_list": [f"tag-{i}" for i in range(100)],
        },
        "model": {
            "gordo_amd.model.anomaly.diff.DiffBasedAnomalyDetector": {
                "base_estimator": {
                    "sklearn.pipeline.Pipeline": {
                        "steps": [
                            "sklearn.preprocessing.MinMaxScaler",
                            {
                                "gordo_amd.model.models.TorchAutoEncoder": {
                                    "kind": "feedforward_hourglass",
                                    "encoding_layers": 2,
                                    "epochs": 2,
                                    "batch_size": 512,
                                }
                            },
                        ]
                    }
                }
            }
        },
        "evaluation": {"cv_mode": "no_cv"},
    }
    out = tmp_path / "wide-serving"
    ModelBuilder(Machine.from_config(config)).build(output_dir=str(out))
    rt = ModelRuntime(str(out))
    assert rt._fused is not None, "wide fused serving path not active"
    assert rt._fused.p.wmax == 128
    rng = np.random.default_rng(0)
    X = pd.DataFrame(
        rng.uniform(0, 1, size=(200, 100)).astype(np.float32),
        columns=[f"tag-{i}" for i in range(100)],
    )
    df = rt.anomaly(X, None)
    tot = df[("total-anomaly-score", "")].values
    assert np.isfinite(tot).all() and (tot >= 0).all()
