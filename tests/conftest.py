import numpy as np
import pandas as pd
import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a real MI355X GPU (run via gpurun)"
    )


@pytest.fixture
def sensor_frame():
    """Small deterministic multivariate frame (10 tags, 200 rows)."""
    rng = np.random.default_rng(42)
    index = pd.date_range("2020-01-01", periods=200, freq="10min", tz="UTC")
    data = np.cumsum(rng.standard_normal((200, 10)) * 0.1, axis=0) + rng.normal(
        0, 1, size=(1, 10)
    )
    return pd.DataFrame(data, index=index, columns=[f"tag-{i}" for i in range(10)])


@pytest.fixture
def machine_config():
    return {
        "name": "test-machine",
        "dataset": {
            "type": "RandomDataset",
            "train_start_date": "2020-01-01T00:00:00Z",
            "train_end_date": "2020-03-01T00:00:00Z",
            "tag_list": [f"tag-{i}" for i in range(10)],
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
                                    "epochs": 2,
                                    "batch_size": 64,
                                }
                            },
                        ]
                    }
                }
            }
        },
    }


@pytest.fixture(scope="session")
def built_model_dir(tmp_path_factory):
    """A real (tiny) anomaly model artifact dir, built once per session."""
    from gordo_amd.builder import ModelBuilder
    from gordo_amd.machine import Machine

    config = {
        "name": "test-machine",
        "dataset": {
            "type": "RandomDataset",
            "train_start_date": "2020-01-01T00:00:00Z",
            "train_end_date": "2020-02-01T00:00:00Z",
            "tag_list": [f"tag-{i}" for i in range(10)],
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
                                    "epochs": 1,
                                    "batch_size": 64,
                                }
                            },
                        ]
                    }
                }
            }
        },
    }
    out = tmp_path_factory.mktemp("model") / "test-machine"
    machine = Machine.from_config(config)
    ModelBuilder(machine).build(output_dir=str(out))
    return str(out)
