"""Client integration tests against a live in-process uvicorn server
(parity: tests/.../test_client.py)."""
import socket
import threading
import time

import numpy as np
import pandas as pd
import pytest


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.fixture(scope="module")
def live_server(built_model_dir):
    import uvicorn

    from gordo_amd.server import build_app

    app = build_app(built_model_dir)
    port = _free_port()
    config = uvicorn.Config(app, host="127.0.0.1", port=port, log_level="warning")
    server = uvicorn.Server(config)
    thread = threading.Thread(target=server.run, daemon=True)
    thread.start()
    for _ in range(100):
        if server.started:
            break
        time.sleep(0.1)
    assert server.started
    yield f"http://127.0.0.1:{port}"
    server.should_exit = True
    thread.join(5)


def test_client_metadata(live_server):
    from gordo_amd.client import Client

    c = Client(endpoints=[live_server])
    meta = c.get_metadata()
    assert "test-machine" in meta


def test_client_predict_with_provider(live_server, tmp_path):
    """Full pull: the client assembles X itself via the dataset layer, posts
    chunked anomaly requests (parquet), reassembles the frame, and forwards
    to disk."""
    from gordo_amd.client import Client, ForwardPredictionsToDisk
    from gordo_amd.dataset.data_provider import RandomDataProvider

    out_dir = tmp_path / "fwd"
    c = Client(
        endpoints=[live_server],
        data_provider=RandomDataProvider(min_size=300, max_size=300),
        prediction_forwarder=ForwardPredictionsToDisk(str(out_dir)),
        max_request_hours=4,
    )
    results = c.predict("2020-01-01T00:00:00Z", "2020-01-01T12:00:00Z")
    assert len(results) == 1
    res = results[0]
    assert res.error_messages == []
    assert res.predictions is not None and len(res.predictions) > 0
    tops = set(res.predictions.columns.get_level_values(0))
    assert "total-anomaly-score" in tops
    assert (out_dir / "test-machine.parquet").exists()


def test_client_predict_with_X(live_server):
    from gordo_amd.client import Client

    rng = np.random.default_rng(0)
    idx = pd.date_range("2020-01-01", periods=50, freq="10min", tz="UTC")
    X = pd.DataFrame(rng.normal(size=(50, 10)), index=idx,
                     columns=[f"tag-{i}" for i in range(10)])
    c = Client(endpoints=[live_server], use_parquet=False)
    results = c.predict(idx[0], idx[-1] + pd.Timedelta("10min"), X=X)
    assert results[0].predictions is not None
    assert len(results[0].predictions) == 50


def test_client_download_model(live_server):
    from gordo_amd.client import Client

    c = Client(endpoints=[live_server])
    models = c.download_model()
    assert "test-machine" in models
    assert hasattr(models["test-machine"], "anomaly")


def test_make_date_ranges():
    from gordo_amd.client import make_date_ranges

    start = pd.Timestamp("2020-01-01", tz="UTC")
    end = pd.Timestamp("2020-01-08", tz="UTC")
    ranges = make_date_ranges(start, end, pd.Timedelta("1D"))
    assert len(ranges) == 7
    assert ranges[0][0] == start and ranges[-1][1] == end


def test_watchman_and_client_discovery(live_server):
    """Watchman polls the live server; the client discovers through it."""
    import uvicorn

    from gordo_amd.client import Client
    from gordo_amd.watchman import build_app as build_watchman

    wapp = build_watchman("proj", {"test-machine": live_server}, refresh_interval=1)
    port = _free_port()
    config = uvicorn.Config(wapp, host="127.0.0.1", port=port, log_level="warning")
    server = uvicorn.Server(config)
    thread = threading.Thread(target=server.run, daemon=True)
    thread.start()
    for _ in range(100):
        if server.started:
            break
        time.sleep(0.1)
    try:
        deadline = time.time() + 15
        healthy = False
        c = Client(watchman_url=f"http://127.0.0.1:{port}/")
        while time.time() < deadline:
            try:
                meta = c.get_metadata()
                if "test-machine" in meta:
                    healthy = True
                    break
            except Exception:
                pass
            time.sleep(0.5)
        assert healthy
    finally:
        server.should_exit = True
        thread.join(5)
