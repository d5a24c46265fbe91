"""Semi-fused wide-LSTM engine (widths beyond the hand-written kernels'
128-feature envelope... in practice any width the HIP cell kernels reject).

torch.nn.LSTM's eager path on ROCm measured ~4K windows/s at hidden 128
(launch-storm-bound); this runs each layer's time loop as ONE rocBLAS GEMM
per step on the concatenated [x_t | h] input against the packed 4-gate
weight, with autograd providing BPTT. Library GEMMs are efficient at these
widths (the hand-written MFMA kernels matter most at tiny widths), so this
is the right MI355X shape for wide models: ~2 orders of magnitude over
nn.LSTM while keeping exact torch semantics.

Weights stay the canonical torch nn.LSTM parameters (state_dict identical;
serializer artifacts unchanged); the packed views are rebuilt per fit.
"""
from __future__ import annotations

import logging
import math
import time
from typing import List

import numpy as np
import torch

from ..runtime.trainer import TrainHistory

logger = logging.getLogger(__name__)


def _packed_weights(module, device, dtype):
    """Per layer: WcatT [I+H, 4H] and bias [4H] built FROM the live torch
    parameters (so autograd grads flow back into the module)."""
    packed = []
    for cell in module.cells:
        wcat_t = torch.cat([cell.weight_ih_l0, cell.weight_hh_l0], dim=1).t()
        b = cell.bias_ih_l0 + cell.bias_hh_l0
        packed.append((wcat_t.to(device=device, dtype=dtype), b.to(device=device, dtype=dtype)))
    return packed


def _lstm_layer(x, wcat_t, b, H):
    """x [B,T,I] -> h_seq [B,T,H] via one addmm per timestep."""
    B, T, _ = x.shape
    h = x.new_zeros(B, H)
    c = x.new_zeros(B, H)
    outs = []
    for t in range(T):
        gates = torch.addmm(b, torch.cat([x[:, t, :], h], dim=1), wcat_t)
        i, f, g, o = gates.split(H, dim=1)
        i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
        g = torch.tanh(g)
        c = f * c + i * g
        h = o * torch.tanh(c)
        outs.append(h)
    return torch.stack(outs, dim=1)


def wide_forward(module, x):
    """Full stacked forward: windows [B,T,F] -> head output [B,F_out]."""
    dtype = x.dtype
    packed = _packed_weights(module, x.device, dtype)
    h = x
    for (wcat_t, b), cell in zip(packed, module.cells):
        h = _lstm_layer(h, wcat_t, b, cell.hidden_size)
    last = h[:, -1, :]
    return module.head_act(module.head(last))


class WideLSTMTrainer:
    """GPU trainer for LSTMNet widths the fused HIP cells don't cover."""

    def __init__(self, module, device: torch.device, cfg):
        self.module = module
        self.device = device
        self.cfg = cfg

    def fit(self, X: np.ndarray, y: np.ndarray, lookback: int = 144, lookahead: int = 0) -> TrainHistory:
        cfg = self.cfg
        device = self.device
        if cfg.seed is not None:
            torch.manual_seed(cfg.seed)
        module = self.module.to(device)
        T = int(lookback)
        Xt = torch.as_tensor(np.asarray(X, np.float32), device=device)
        y_is_x = y is X or (X.shape == y.shape and np.shares_memory(X, y))
        yt = Xt if y_is_x else torch.as_tensor(np.asarray(y, np.float32), device=device)
        n_windows = len(Xt) - T + 1 - lookahead
        if n_windows <= 0:
            raise ValueError(f"need > {T + lookahead} rows for lookback={T}")
        B = max(1, min(cfg.batch_size, n_windows))
        n_batches = max(1, math.ceil(n_windows / B))
        offsets = torch.arange(T, device=device)

        opt = torch.optim.Adam(module.parameters(), lr=cfg.lr, betas=cfg.betas,
                               eps=cfg.eps, foreach=True)
        loss_fn = torch.nn.MSELoss()
        use_bf16 = device.type == "cuda" and cfg.dtype == "bf16"

        history = TrainHistory(engine="wide_lstm_gemmloop")
        torch.cuda.synchronize() if device.type == "cuda" else None
        t0 = time.perf_counter()
        for _ in range(cfg.epochs):
            perm = torch.randperm(n_windows, device=device)
            total = 0.0
            for s in range(n_batches):
                idx = perm[s * B : (s + 1) * B]
                w = Xt[idx.unsqueeze(1) + offsets.unsqueeze(0)]
                tgt = yt[idx + T - 1 + lookahead]
                opt.zero_grad(set_to_none=True)
                if use_bf16:
                    with torch.autocast("cuda", dtype=torch.bfloat16):
                        out = wide_forward(module, w)
                    loss = loss_fn(out.float(), tgt)
                else:
                    loss = loss_fn(wide_forward(module, w), tgt)
                loss.backward()
                opt.step()
                total += float(loss.detach()) * len(idx)
            history.loss.append(total / (n_batches * B))
        if device.type == "cuda":
            torch.cuda.synchronize()
        history.duration_sec = time.perf_counter() - t0
        history.epochs = cfg.epochs
        history.samples_per_sec = (
            cfg.epochs * n_batches * B / history.duration_sec if history.duration_sec else 0.0
        )
        return history
