"""Soak the fused reduce+Adam step path: 100K steps through the multi-step
replay tiers, then verify (a) the ping-pong counters hold exactly N steps of
state, (b) parameters are finite, (c) the loss over the last epoch-window
is below the first (still learning / not diverged).

Run on a GPU box:  python tools/soak_fused_ra.py [n_steps]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from gordo_amd.model.factories import feedforward_hourglass
from gordo_amd.parallel.ddp import DPContext, HourglassStepTrainer
from gordo_amd.runtime.trainer import TrainConfig


def main(n_steps: int = 100_000):
    torch.manual_seed(0)
    rng = np.random.default_rng(0)
    Z = rng.normal(size=(65536, 8)).astype(np.float32)
    P = rng.normal(size=(8, 50)).astype(np.float32)
    X = np.tanh(Z @ P)

    module = feedforward_hourglass(50, 50, encoding_layers=3, compression_factor=0.5)
    cfg = TrainConfig(batch_size=8192, lr=1e-3, dtype="bf16")
    tr = HourglassStepTrainer(module, X, 8192, cfg, dp=DPContext(device=torch.device("cuda:0")))
    assert tr.fused and tr._graphs is not None and tr._multi_graphs

    tr.steps(64)
    tr.synchronize()
    loss_head = tr.epoch_loss() / 64  # accumulated since last drain

    t0 = time.perf_counter()
    tr.steps(n_steps - 128)
    tr.synchronize()
    dt = time.perf_counter() - t0
    tr.epoch_loss()  # drain the middle window
    tr.steps(64)
    tr.synchronize()

    live = tr._cnt[tr._parity].cpu().tolist()
    nb = tr.n_batches
    assert live == [n_steps % nb, n_steps, n_steps // nb], live
    vec = tr.params_vector().cpu().numpy()
    assert np.isfinite(vec).all()
    loss_tail = tr.epoch_loss() / 64
    sps = (n_steps - 128) * 8192 / dt
    print(
        f"soak ok: {n_steps} steps, {sps/1e6:.1f}M samples/s sustained, "
        f"counters {live}, loss head->tail {loss_head:.4g}->{loss_tail:.4g}, "
        f"params finite"
    )
    assert loss_tail < loss_head


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 100_000)
