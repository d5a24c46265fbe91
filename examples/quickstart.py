"""End-to-end example: YAML project -> fleet build -> load -> anomaly scores.

Run: python examples/quickstart.py  (CPU works; MI355X uses the fused kernels)
"""
import os
import sys
import tempfile

import numpy as np
import pandas as pd
import yaml

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from gordo_amd.machine import NormalizedConfig
from gordo_amd.server.model_io import ModelRuntime
from gordo_amd.workflow.fleet import FleetBuilder


def main():
    with open(os.path.join(os.path.dirname(__file__), "project.yaml")) as f:
        config = NormalizedConfig(yaml.safe_load(f))

    out_dir = tempfile.mkdtemp(prefix="gordo-example-")
    results = FleetBuilder(config, output_dir=out_dir).build_all()
    for r in results:
        print(f"{r['machine']:14s} {r['status']:9s} {r['seconds']:.1f}s")

    # score fresh data against the first machine's model
    rt = ModelRuntime(os.path.join(out_dir, "compressor-a"))
    rng = np.random.default_rng(0)
    X = pd.DataFrame(
        rng.uniform(0, 1, size=(500, 10)).astype(np.float32),
        columns=[f"tag-{i}" for i in range(10)],
        index=pd.date_range("2021-01-01", periods=500, freq="10min"),
    )
    scores = rt.anomaly(X, None)
    print(scores[("total-anomaly-score", "")].describe())


if __name__ == "__main__":
    main()
