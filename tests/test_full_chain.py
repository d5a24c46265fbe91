"""Capstone integration test: the WHOLE framework end to end on CPU.

project YAML → NormalizedConfig → FleetBuilder (per-machine artifacts with
metadata + thresholds) → fleet server (every machine mounted under
/gordo/v0/<project>/<machine>) → async Client (discovery via the fleet
root, chunked anomaly pulls) → disk forwarder. This is the reference's
full deployment story (YAML → builder pods → model servers →
watchman → client → Influx) collapsed onto one node.
"""
import socket
import threading
import time

import numpy as np
import pandas as pd
import pytest

N_TAGS = 6
TAGS = [f"tag-{i}" for i in range(N_TAGS)]


def _project(tmp_path):
    return {
        "project_name": "chain",
        "machines": [
            {"name": "m-a"},
            {"name": "m-b", "dataset": {"tag_list": TAGS[:4]}},
        ],
        "globals": {
            "dataset": {
                "type": "RandomDataset",
                "train_start_date": "2020-01-01T00:00:00Z",
                "train_end_date": "2020-03-01T00:00:00Z",
                "tag_list": TAGS,
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
                                        "seed": 0,
                                    }
                                },
                            ]
                        }
                    }
                }
            },
            "evaluation": {"cv_mode": "full_build", "cv_splits": 2},
        },
    }


@pytest.mark.timeout(300)
def test_full_chain_config_to_forwarded_predictions(tmp_path):
    from gordo_amd.client import Client
    from gordo_amd.client.forwarders import ForwardPredictionsToDisk
    from gordo_amd.machine import NormalizedConfig
    from gordo_amd.serializer import load_metadata
    from gordo_amd.server.fleet_app import build_fleet_app
    from gordo_amd.workflow.fleet import FleetBuilder

    # ---- 1. config -> built fleet ------------------------------------
    cfg = NormalizedConfig(_project(tmp_path))
    out_dir = tmp_path / "models"
    results = FleetBuilder(cfg, output_dir=str(out_dir), n_workers=2).build_all()
    assert all(r["status"] == "succeeded" for r in results)

    meta = load_metadata(str(out_dir / "m-a"))
    assert meta["name"] == "m-a"
    cvm = meta["model"]["cross-validation"]
    assert "explained_variance_score" in cvm["scores"]
    assert "thresholds" in cvm  # anomaly thresholds derived during CV

    # ---- 2. fleet server over every artifact -------------------------
    import uvicorn

    app = build_fleet_app(str(out_dir), project_name="chain")
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    server = uvicorn.Server(
        uvicorn.Config(app, host="127.0.0.1", port=port, log_level="warning")
    )
    thread = threading.Thread(target=server.run, daemon=True)
    thread.start()
    for _ in range(100):
        if server.started:
            break
        time.sleep(0.1)
    assert server.started

    try:
        # ---- 3. async client: discovery + chunked anomaly pull --------
        forward_dir = tmp_path / "forwarded"
        client = Client(
            project="chain",
            watchman_url=f"http://127.0.0.1:{port}/",
            prediction_forwarder=ForwardPredictionsToDisk(str(forward_dir)),
        )
        rng = np.random.default_rng(0)
        idx = pd.date_range("2021-01-01", periods=48, freq="10min", tz="UTC")
        X = pd.DataFrame(
            rng.uniform(0, 1, size=(48, N_TAGS)).astype(np.float32),
            columns=TAGS,
            index=idx,
        )
        results = client.predict(idx[0], idx[-1] + pd.Timedelta("10min"), X=X)
        assert sorted(r.name for r in results) == ["m-a", "m-b"]
        for r in results:
            assert not r.error_messages, r.error_messages
            tops = set(r.predictions.columns.get_level_values(0))
            assert "total-anomaly-score" in tops
            # m-b trained on 4 tags: its frame has 4 model-output columns
            width = len(r.predictions["model-output"].columns)
            assert width == (4 if r.name == "m-b" else N_TAGS)

        # ---- 4. forwarder wrote per-machine parquet -------------------
        from gordo_amd.server.utils import dataframe_from_parquet_bytes

        for name in ("m-a", "m-b"):
            blob = (forward_dir / f"{name}.parquet").read_bytes()
            df = dataframe_from_parquet_bytes(blob)
            assert "total-anomaly-score" in set(df.columns.get_level_values(0))
            assert len(df) == 48
    finally:
        server.should_exit = True
        thread.join(5)
