#!/usr/bin/env python3
"""Fleet benchmark (BASELINE config #5): build N per-machine anomaly models
through the fleet builder, then measure anomaly-score latency through the
serving runtime (the fused scale→forward→|err|→L2 HIP kernel per model).

  python bench_serving.py --machines 32 --requests 500 --req-batch 1000

Prints one JSON line: models built/hour + anomaly p50/p99 latency.
(Single-node; the driver's headline bench stays bench.py.)
"""
import argparse
import json
import os
import sys
import tempfile
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

N_TAGS = 50
EVALUATION = {"cv_mode": "no_cv"}


def project_config(n_machines: int, n_tags: int):
    machines = []
    for i in range(n_machines):
        machines.append({"name": f"machine-{i:03d}"})
    return {
        "project_name": "bench-fleet",
        "machines": machines,
        "globals": {
            "dataset": {
                "type": "RandomDataset",
                "train_start_date": "2020-01-01T00:00:00Z",
                "train_end_date": "2020-06-01T00:00:00Z",
                "tag_list": [f"tag-{j}" for j in range(n_tags)],
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
                                        "compression_factor": 0.5,
                                        "epochs": 10,
                                        "batch_size": 4096,
                                    }
                                },
                            ]
                        }
                    }
                }
            },
            "evaluation": EVALUATION,
        },
    }


def measure_http(out_dir: str, args, X):
    """End-to-end HTTP: uvicorn fleet server in a subprocess; measures both
    wire formats (parquet for reference parity, arrow IPC, raw float32 = the
    fast paths)."""
    import socket
    import subprocess
    import httpx
    from gordo_amd.server.utils import (
        dataframe_into_arrow_bytes,
        dataframe_into_parquet_bytes,
    )

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    repo = os.path.dirname(os.path.abspath(__file__))
    env = dict(os.environ, PYTHONPATH=repo + os.pathsep + os.environ.get("PYTHONPATH", ""))
    proc = subprocess.Popen(
        [sys.executable, "-m", "gordo_amd.cli.cli", "fleet", "serve",
         "--models-dir", out_dir, "--project-name", "bench-fleet",
         "--host", "127.0.0.1", "--port", str(port)],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL, env=env,
    )
    try:
        base = f"http://127.0.0.1:{port}"
        body = dataframe_into_parquet_bytes(X)
        with httpx.Client(timeout=30.0) as client:
            for _ in range(240):  # wait for startup (model loads)
                try:
                    if client.get(base + "/healthcheck").status_code == 200:
                        break
                except httpx.TransportError:
                    pass
                time.sleep(0.5)
            out = {}
            abody = dataframe_into_arrow_bytes(X)
            rarr = np.ascontiguousarray(X.to_numpy(np.float32))
            rheaders = {"content-type": "application/x-gordo-raw",
                        "X-Gordo-Shape": f"{rarr.shape[0]},{rarr.shape[1]}"}
            for fmt, payload, headers in (
                ("parquet", body, {"content-type": "application/octet-stream"}),
                ("arrow", abody,
                 {"content-type": "application/vnd.apache.arrow.stream"}),
                ("raw", rarr.tobytes(), rheaders),
            ):
                urls = [
                    f"{base}/gordo/v0/bench-fleet/machine-{i:03d}/anomaly/prediction?format={fmt}"
                    for i in range(min(8, args.machines))
                ]
                for u in urls:  # warm
                    r = client.post(u, content=payload, headers=headers)
                    assert r.status_code == 200, r.text[:200]
                lat = []
                for i in range(min(args.requests, 200)):
                    t = time.perf_counter()
                    r = client.post(urls[i % len(urls)], content=payload,
                                    headers=headers)
                    assert r.status_code == 200
                    lat.append(time.perf_counter() - t)
                lat_ms = np.array(lat) * 1e3
                key = "http_" + ("" if fmt == "parquet" else fmt + "_")
                out[key + "p50_ms"] = float(np.percentile(lat_ms, 50))
                out[key + "p99_ms"] = float(np.percentile(lat_ms, 99))
        return out
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--machines", type=int, default=32)
    ap.add_argument("--tags", type=int, default=N_TAGS)
    ap.add_argument("--requests", type=int, default=500)
    ap.add_argument("--req-batch", type=int, default=1000)
    ap.add_argument("--out-dir", default=None)
    ap.add_argument("--workers", type=int, default=None,
                    help="builder workers (default: one per GPU)")
    ap.add_argument("--http", action="store_true",
                    help="also measure end-to-end HTTP latency through the fleet server")
    ap.add_argument("--cv", action="store_true",
                    help="full_build evaluation: 3-fold TimeSeriesSplit CV with "
                    "per-tag metrics + anomaly thresholds per machine (the "
                    "torch-vectorized K9 path) instead of no_cv")
    args = ap.parse_args()
    if args.cv:
        EVALUATION.clear()
        EVALUATION.update({
            "cv_mode": "full_build",
            "cv_splits": 3,
            "metrics": [
                "explained_variance_score", "r2_score",
                "mean_squared_error", "mean_absolute_error",
            ],
            "per_tag_metrics": True,
        })

    from gordo_amd.machine import NormalizedConfig
    from gordo_amd.server.model_io import ModelRuntime
    from gordo_amd.workflow.fleet import FleetBuilder

    out_dir = args.out_dir or tempfile.mkdtemp(prefix="gordo-fleet-")
    cfg = NormalizedConfig(project_config(args.machines, args.tags))

    t0 = time.perf_counter()
    builder = FleetBuilder(cfg, output_dir=out_dir, n_workers=args.workers)
    results = builder.build_all()
    build_s = time.perf_counter() - t0
    ok = sum(1 for r in results if r["status"] == "succeeded")
    assert ok == args.machines, [r for r in results if r["status"] != "succeeded"][:1]

    # ---- serving latency: fused anomaly path, one runtime per machine -----
    # (the runtime loop is the fleet COLD LOAD: unpickle object graph +
    # mmap weights from weights.safetensors + build the fused forward)
    t0 = time.perf_counter()
    runtimes = [
        ModelRuntime(os.path.join(out_dir, f"machine-{i:03d}"))
        for i in range(args.machines)
    ]
    cold_load_s = time.perf_counter() - t0
    import pandas as pd

    rng = np.random.default_rng(0)
    tags = [f"tag-{j}" for j in range(args.tags)]
    X = pd.DataFrame(
        rng.uniform(0, 1, size=(args.req_batch, args.tags)).astype(np.float32),
        columns=tags,
        index=pd.date_range("2021-01-01", periods=args.req_batch, freq="10min"),
    )
    # warmup
    for rt in runtimes[: min(8, len(runtimes))]:
        rt.anomaly(X, None)
    lat = []
    for i in range(args.requests):
        rt = runtimes[i % len(runtimes)]
        t = time.perf_counter()
        df = rt.anomaly(X, None)
        lat.append(time.perf_counter() - t)
    assert "total-anomaly-score" in set(c[0] for c in df.columns)
    lat_ms = np.array(lat) * 1e3
    fused = sum(1 for rt in runtimes if rt._fused is not None)

    http_stats = {}
    if args.http:
        http_stats = measure_http(out_dir, args, X)

    print(json.dumps({
        **http_stats,
        "metric": "anomaly_p50_latency_ms",
        "value": float(np.percentile(lat_ms, 50)),
        "unit": "ms",
        "p99_ms": float(np.percentile(lat_ms, 99)),
        "req_batch_rows": args.req_batch,
        "n_models": args.machines,
        "fused_runtimes": fused,
        "models_built": ok,
        "build_seconds": build_s,
        "models_per_hour": ok / build_s * 3600,
        "cv": bool(args.cv),
        "cold_load_seconds": cold_load_s,
        "cold_load_ms_per_model": cold_load_s / args.machines * 1e3,
        "higher_is_better": False,
        "data": "synthetic",
    }))


if __name__ == "__main__":
    main()
