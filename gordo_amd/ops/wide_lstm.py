"""Semi-fused wide-LSTM engine (widths the hand-written HIP cells don't
cover; any hidden size).

torch.nn.LSTM's eager path on ROCm measured ~4K windows/s at hidden 128
with a 144-step lookback. This engine batches the input projection for ALL
timesteps into one rocBLAS GEMM per layer and captures the whole
forward+backward in a hipGraph replayed per batch (static shapes) —
measured 2.5x nn.LSTM. The remaining ceiling is per-kernel execution cost
of the ~7000 small captured kernels in the 144-step autograd chain;
closing it needs the persistent fused cell kernels generalized past
64-wide (docs/ROADMAP.md).

Weights stay the canonical torch nn.LSTM parameters (state_dict identical;
serializer artifacts unchanged).
"""
from __future__ import annotations

import logging
import math
import os
import time
import numpy as np
import torch

from ..runtime.trainer import TrainHistory

logger = logging.getLogger(__name__)


def _packed_weights(module, device, dtype):
    """Per layer: (W_ih^T [I,4H], W_hh^T [H,4H], bias [4H]) built FROM the
    live torch parameters (autograd grads flow back into the module)."""
    packed = []
    for cell in module.cells:
        w_ih_t = cell.weight_ih_l0.t().to(device=device, dtype=dtype)
        w_hh_t = cell.weight_hh_l0.t().to(device=device, dtype=dtype)
        b = (cell.bias_ih_l0 + cell.bias_hh_l0).to(device=device, dtype=dtype)
        packed.append((w_ih_t, w_hh_t, b))
    return packed


def _lstm_layer(x, w_ih_t, w_hh_t, b, H):
    """x [B,T,I] -> h_seq [B,T,H]: one x-projection GEMM for all timesteps,
    then per step one mm + ATen's fused differentiable LSTM cell (the same
    pointwise kernel torch.nn.LSTM uses on CUDA; dispatching it directly
    bypasses the slow MIOpen RNN route on ROCm)."""
    B, T, I = x.shape
    # [T,B,4H] so each timestep slice is contiguous for the fused cell
    gx = (
        torch.addmm(b, x.reshape(-1, I), w_ih_t)
        .view(B, T, 4 * H).transpose(0, 1).contiguous()
    )
    h = x.new_zeros(B, H)
    c = x.new_zeros(B, H)
    outs = []
    if x.is_cuda:
        for t in range(T):
            gh = h.mm(w_hh_t)
            h, c, _ = torch.ops.aten._thnn_fused_lstm_cell(gx[t], gh, c)
            outs.append(h)
    else:
        for t in range(T):
            gates = torch.addmm(gx[t], h, w_hh_t)
            i, f, g, o = gates.split(H, dim=1)
            i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
            g = torch.tanh(g)
            c = f * c + i * g
            h = o * torch.tanh(c)
            outs.append(h)
    return torch.stack(outs, dim=1)


def wide_forward(module, x):
    """Full stacked forward: windows [B,T,F] -> head output [B,F_out]."""
    packed = _packed_weights(module, x.device, x.dtype)
    h = x
    for (w_ih_t, w_hh_t, b), cell in zip(packed, module.cells):
        h = _lstm_layer(h, w_ih_t, w_hh_t, b, cell.hidden_size)
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
        F_in = Xt.shape[1]
        F_out = module.n_features_out

        opt = torch.optim.Adam(module.parameters(), lr=cfg.lr, betas=cfg.betas,
                               eps=cfg.eps, foreach=True)
        loss_fn = torch.nn.MSELoss()

        # fp32 throughout: the step is kernel-count-bound, not GEMM-bound,
        # and the fused ATen cell is fp32-first
        def fwd_loss(w, tgt):
            return loss_fn(wide_forward(module, w), tgt)

        # ---- hipGraph capture of the whole fwd+bwd (static shapes) --------
        graph = None
        static_w = static_tgt = static_loss = None
        if (device.type == "cuda" and getattr(cfg, "use_graph", True)
                and not os.environ.get("GORDO_NO_GRAPH")):
            try:
                static_w = torch.zeros(B, T, F_in, device=device)
                static_tgt = torch.zeros(B, F_out, device=device)
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    for _ in range(2):  # warmup: allocator + autocast caches
                        opt.zero_grad(set_to_none=True)
                        fwd_loss(static_w, static_tgt).backward()
                torch.cuda.current_stream().wait_stream(s)
                torch.cuda.synchronize()
                opt.zero_grad(set_to_none=True)
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    static_loss = fwd_loss(static_w, static_tgt)
                    static_loss.backward()
                graph = g
                # reset weights perturbed by nothing (no opt.step during
                # warmup), but grads are stale: zero them for the real run
                for p in module.parameters():
                    if p.grad is not None:
                        p.grad.zero_()
                torch.cuda.synchronize()
            except Exception as e:
                logger.warning("wide-LSTM graph capture failed (%s); eager loop", e)
                graph = None

        history = TrainHistory(
            engine="wide_lstm_gemmloop" + ("_graph" if graph else "")
        )
        if device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(cfg.epochs):
            perm = torch.randperm(n_windows, device=device)
            if n_batches * B > n_windows:  # pad so every batch is exactly B
                extra = torch.randint(0, n_windows, (n_batches * B - n_windows,), device=device)
                perm = torch.cat([perm, extra])
            total = 0.0
            for s_i in range(n_batches):
                idx = perm[s_i * B : (s_i + 1) * B]
                if graph is not None:
                    torch.index_select(Xt, 0, (idx.unsqueeze(1) + offsets.unsqueeze(0)).reshape(-1),
                                       out=static_w.view(B * T, F_in))
                    torch.index_select(yt, 0, idx + T - 1 + lookahead, out=static_tgt)
                    graph.replay()
                    opt.step()
                    opt.zero_grad(set_to_none=False)  # keep graph's grad buffers
                    total += float(static_loss.detach()) * B
                else:
                    w = Xt[idx.unsqueeze(1) + offsets.unsqueeze(0)]
                    tgt = yt[idx + T - 1 + lookahead]
                    opt.zero_grad(set_to_none=True)
                    loss = fwd_loss(w, tgt)
                    loss.backward()
                    opt.step()
                    total += float(loss.detach()) * B
            history.loss.append(total / (n_batches * B))
        if device.type == "cuda":
            torch.cuda.synchronize()
        history.duration_sec = time.perf_counter() - t0
        history.epochs = cfg.epochs
        history.samples_per_sec = (
            cfg.epochs * n_batches * B / history.duration_sec if history.duration_sec else 0.0
        )
        return history
