"""Fused hourglass training/inference engine (Python driver for the CDNA4
kernels in hip/hourglass_kernels.hip).

Training step = 3 device-side kernels (train_step → adam → bump) with zero
host round-trips, captured into one hipGraph (torch.cuda.CUDAGraph) and
replayed per step; loss is read back once per epoch.
"""
from __future__ import annotations

import logging
import math
import os
import time
from typing import List, Optional

import numpy as np
import torch

from . import ext
from ..runtime.trainer import TrainHistory

logger = logging.getLogger(__name__)

WMAX = 64         # narrow-variant padded width (also the LSTM head width)
WMAX_WIDE = 128   # wide-variant padded width
WMAX_MAX = 256    # widest instantiation (> this -> library-GEMM fallback)
MAXL = 8          # kernel arg-array cap (common.h GA_MAXL)
HG_MT = 32        # batch rows per block tile


def max_layers_for(wmax: int) -> int:
    """Per-width fused layer limit: (L+4) activation/scratch tiles + the
    bias image must fit the 160 KiB LDS. 8 at W=64/128, 5 at W=256 —
    covers the factory DEFAULT hourglass (encoding_layers=3 -> L=7 with
    the duplicated-bottleneck composition) on <=128-wide models."""
    tb = HG_MT * wmax * 2
    budget = 160 * 1024 - MAXL * wmax * 4 - 1024
    return max(1, min(MAXL, budget // tb - 4))
ACT_IDS = {"linear": 0, None: 0, "tanh": 1, "relu": 2, "sigmoid": 3}


def wmax_for(widths) -> int:
    """Padded kernel width (64/128/256 instantiation tier) for a layer stack."""
    w = max(widths)
    if w <= WMAX:
        return WMAX
    return WMAX_WIDE if w <= WMAX_WIDE else WMAX_MAX


def supported(module) -> Optional[str]:
    """None if the fused path supports this FeedForwardNet, else the reason."""
    widths = module.layer_widths
    if max(widths) > WMAX_MAX:
        return f"width {max(widths)} > {WMAX_MAX}"
    L = len(widths) - 1
    lmax = max_layers_for(wmax_for(widths))
    if L < 1 or L > lmax:
        return f"L={L} outside 1..{lmax} (width tier {wmax_for(widths)})"
    for f in module.layer_funcs:
        if f not in ACT_IDS:
            return f"activation {f!r} not fused"
    return None


class _PackedParams:
    """Padded device-side parameter state shared by train + inference."""

    def __init__(self, module, device):
        self.widths = module.layer_widths
        self.L = len(self.widths) - 1
        self.act_ids: List[int] = [ACT_IDS[f] for f in module.layer_funcs]
        self.device = device
        WM = wmax_for(self.widths)
        self.wmax = WM
        PW = self.L * WM * WM
        self.PW = PW
        self.P = PW + self.L * WM

        master = torch.zeros(self.P, dtype=torch.float32, device=device)
        linears = [m for m in module.net if isinstance(m, torch.nn.Linear)]
        assert len(linears) == self.L
        self.linears = linears
        for i, lin in enumerate(linears):
            out_w, in_w = lin.weight.shape
            wpad = master[i * WM * WM : (i + 1) * WM * WM].view(WM, WM)
            wpad[:out_w, :in_w] = lin.weight.detach().to(device, torch.float32)
            bpad = master[PW + i * WM : PW + (i + 1) * WM]
            bpad[:out_w] = lin.bias.detach().to(device, torch.float32)
        self.master = master
        self.bias = master[PW:].clone()  # fp32 bias view used by kernels
        w = master[:PW].view(self.L, WM, WM)
        self.w_bf16 = w.to(torch.bfloat16).contiguous()
        self.wt_bf16 = w.transpose(1, 2).contiguous().to(torch.bfloat16).contiguous()

    def sync_to_module(self):
        """Copy trained padded weights back into the torch module (fp32)."""
        WM = self.wmax
        w = self.master[: self.PW].view(self.L, WM, WM)
        b = self.master[self.PW :].view(self.L, WM)
        with torch.no_grad():
            for i, lin in enumerate(self.linears):
                out_w, in_w = lin.weight.shape
                lin.weight.copy_(w[i, :out_w, :in_w].to(lin.weight.device))
                lin.bias.copy_(b[i, :out_w].to(lin.bias.device))


def feistel_kbits(n: int) -> int:
    """Even bit-width of the shuffle's Feistel domain for n rows."""
    k = max(2, int(math.ceil(math.log2(max(2, n)))))
    return k + (k & 1)


def pad_rows_bf16(X: np.ndarray, device, wmax: int = WMAX) -> torch.Tensor:
    """[N,F] fp32 → [N,wmax] bf16 device tensor, zero-padded."""
    N, F = X.shape
    t = torch.zeros(N, wmax, dtype=torch.bfloat16, device=device)
    t[:, :F] = torch.as_tensor(X, device=device, dtype=torch.float32).to(torch.bfloat16)
    return t


class FusedHourglassTrainer:
    def __init__(self, module, device: torch.device, cfg):
        reason = supported(module)
        if reason:
            raise ValueError(f"fused path unsupported: {reason}")
        self.module = module
        self.device = device
        self.cfg = cfg
        self.E = ext()
        if self.E is None:
            raise RuntimeError("HIP extension missing")

    def fit(self, X: np.ndarray, y: np.ndarray, lookback: int = 0, lookahead: int = 0) -> TrainHistory:
        if lookback:
            raise ValueError("fused hourglass engine is for 2-D models")
        cfg = self.cfg
        device = self.device
        if cfg.seed is not None:
            torch.manual_seed(cfg.seed)
        p = _PackedParams(self.module, device)
        L, F_in, F_out = p.L, p.widths[0], p.widths[-1]

        y_is_x = y is X or (X.shape == y.shape and np.shares_memory(X, y))
        WM = p.wmax
        Xd = pad_rows_bf16(np.asarray(X, np.float32), device, WM)
        Yd = Xd if y_is_x else pad_rows_bf16(np.asarray(y, np.float32), device, WM)

        N = Xd.shape[0]
        B = max(64, min(cfg.batch_size, N))
        # GORDO_W1=1 opts into the wave-local W=64 step (one 64-lane wave
        # owns 16 rows, zero barriers). Measured SLOWER at the headline
        # config (184.7M vs 263.5M samples/s): B=8192 yields only 512
        # single-wave blocks, so half the SIMDs idle and the solo wave eats
        # its LDS/global latencies — the 4-wave kernel's barrier parking is
        # the parallelism ceiling, not waste (docs/ROADMAP.md). Kept
        # correct + tested for larger-batch experiments.
        use_w1 = (
            WM == 64
            and hasattr(self.E, "hg_train_step_w1")
            and os.environ.get("GORDO_W1") == "1"
        )
        train_step = self.E.hg_train_step_w1 if use_w1 else self.E.hg_train_step
        MT = int(getattr(self.E, "MT_W1", 16)) if use_w1 else int(getattr(self.E, "MT", 32))
        n_tiles = (B + MT - 1) // MT
        n_batches = max(1, math.ceil(N / B))

        # bf16 slabs: halves the dominant dW-slab HBM stream (fp32 masters
        # still accumulate exactly; per-tile partials round at ~bf16 eps)
        dw = torch.zeros(n_tiles, L, WM * WM, dtype=torch.bfloat16, device=device)
        db = torch.zeros(n_tiles, L, WM, dtype=torch.bfloat16, device=device)
        loss_acc = torch.zeros(1, dtype=torch.float32, device=device)
        # counters [batch idx, adam t, epoch], PING-PONG: the fused
        # hg_reduce_adam reads slot q (written by the previous step) and
        # writes the bumped values to slot q^1 — a single tensor would race
        # the bump against the other blocks' reads inside one launch
        cnt = [
            torch.tensor([0, 0, 0], dtype=torch.int32, device=device),
            torch.tensor([0, 0, 0], dtype=torch.int32, device=device),
        ]
        m = torch.zeros_like(p.master)
        v = torch.zeros_like(p.master)

        gscale = 2.0 / (B * F_out)
        acts = [int(i) for i in p.act_ids]
        kbits = feistel_kbits(N)
        seed = int(cfg.seed) if cfg.seed is not None else 0x5EED

        def one_step(q: int):
            train_step(
                Xd, Yd, cnt[q], p.w_bf16, p.wt_bf16, p.bias, dw, db,
                loss_acc, B, F_out, acts, gscale, bool(y_is_x), None, WM,
                kbits, seed, [int(x) for x in p.widths],
            )
            # fused slab reduce + Adam (replaces the slab_reduce/adam_p pair)
            self.E.hg_reduce_adam(
                dw, db, p.master, m, v, p.w_bf16, p.wt_bf16, p.bias,
                cnt[q], cnt[q ^ 1], n_tiles, L, n_batches,
                cfg.lr, cfg.betas[0], cfg.betas[1], cfg.eps, WM,
            )

        def reset_counters():
            cnt[0].copy_(torch.tensor([0, 0, 0], dtype=torch.int32, device=device))

        # ---- optional hipGraph capture: one graph per counter parity -----
        graphs = None
        if (getattr(cfg, "use_graph", True) and n_batches >= 1
                and not os.environ.get("GORDO_NO_GRAPH")):
            try:
                # warm-up both parities on a side stream (allocations + modules)
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    one_step(0)
                    one_step(1)
                torch.cuda.current_stream().wait_stream(s)
                torch.cuda.synchronize()
                # reset state disturbed by warm-up
                reset_counters()
                loss_acc.zero_()
                m.zero_(); v.zero_()
                p2 = _PackedParams(self.module, device)
                p.master.copy_(p2.master); p.bias.copy_(p2.bias)
                p.w_bf16.copy_(p2.w_bf16); p.wt_bf16.copy_(p2.wt_bf16)
                graphs = []
                for q in (0, 1):
                    g = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(g):
                        one_step(q)
                    graphs.append(g)
                # even-parity multi-step graphs: 2/4/8/16 steps per replay —
                # cut the per-step Python/replay overhead in the epoch loop.
                # Only worth the extra captures on jobs with enough steps:
                # each capture costs ~5 ms per fit() while the tiers save
                # ~3.5 us/step — breakeven ~4k steps. Fleet builds (~300
                # steps/model) measured 10.1 -> 13.0 s per 256 models with
                # these captured unconditionally.
                multi_graphs = []
                if cfg.epochs * n_batches >= 4096:
                    for size in (2, 4, 8, 16):
                        g = torch.cuda.CUDAGraph()
                        with torch.cuda.graph(g):
                            for i in range(size):
                                one_step(i & 1)
                        multi_graphs.append((size, g))
                    multi_graphs.reverse()  # widest first for greedy replay
                # capture replays with pristine state below
                reset_counters()
                loss_acc.zero_(); m.zero_(); v.zero_()
                p.master.copy_(p2.master); p.bias.copy_(p2.bias)
                p.w_bf16.copy_(p2.w_bf16); p.wt_bf16.copy_(p2.wt_bf16)
            except Exception as e:
                logger.warning("hipGraph capture failed (%s); plain launches", e)
                graphs = None
                multi_graphs = []
                # warm-up may have stepped the real state before the failure:
                # restore so the raw-launch run still starts pristine
                torch.cuda.synchronize()
                reset_counters()
                loss_acc.zero_(); m.zero_(); v.zero_()
                p2 = _PackedParams(self.module, device)
                p.master.copy_(p2.master); p.bias.copy_(p2.bias)
                p.w_bf16.copy_(p2.w_bf16); p.wt_bf16.copy_(p2.wt_bf16)

        history = TrainHistory(engine="fused_hip" + ("_graph" if graphs else ""))
        denom = float(n_batches) * B * F_out
        parity = 0
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(cfg.epochs):
            r = n_batches
            if graphs is not None:
                while r:
                    if parity == 0:
                        for size, g in multi_graphs:  # widest first; 0 -> 0
                            if r >= size:
                                g.replay()
                                r -= size
                                break
                        else:
                            graphs[0].replay()
                            parity = 1
                            r -= 1
                    else:
                        graphs[1].replay()
                        parity = 0
                        r -= 1
            else:
                while r:
                    one_step(parity)
                    parity ^= 1
                    r -= 1
            history.loss.append(float(loss_acc.item()) / denom)
            loss_acc.zero_()
        torch.cuda.synchronize()
        history.duration_sec = time.perf_counter() - t0
        history.epochs = cfg.epochs
        history.samples_per_sec = (
            cfg.epochs * n_batches * B / history.duration_sec if history.duration_sec else 0.0
        )
        p.sync_to_module()
        return history


class FusedHourglassForward:
    """Serving/predict forward: optional fused input scaling and anomaly."""

    def __init__(self, module, device: torch.device):
        reason = supported(module)
        if reason:
            raise ValueError(f"fused path unsupported: {reason}")
        self.E = ext()
        if self.E is None:
            raise RuntimeError("HIP extension missing")
        self.device = device
        self.p = _PackedParams(module, device)
        self.F_in = self.p.widths[0]
        self.F_out = self.p.widths[-1]
        WM = self.p.wmax
        self.identity_scale = torch.zeros(WM, dtype=torch.float32, device=device)
        self.identity_scale[: self.F_in] = 1.0
        self.zero_shift = torch.zeros(WM, dtype=torch.float32, device=device)

    def set_input_scaler(self, scale: np.ndarray, shift: np.ndarray):
        """x_scaled = x*scale + shift (fused MinMaxScaler)."""
        s = torch.zeros(self.p.wmax, dtype=torch.float32, device=self.device)
        t = torch.zeros(self.p.wmax, dtype=torch.float32, device=self.device)
        s[: len(scale)] = torch.as_tensor(scale, dtype=torch.float32, device=self.device)
        t[: len(shift)] = torch.as_tensor(shift, dtype=torch.float32, device=self.device)
        self.identity_scale = s
        self.zero_shift = t

    def forward(self, X: torch.Tensor) -> torch.Tensor:
        X = X.to(self.device, torch.float32).contiguous()
        yhat = torch.empty(X.shape[0], self.F_out, dtype=torch.float32, device=self.device)
        self.E.hg_forward(
            X, self.p.w_bf16, self.p.bias, self.identity_scale, self.zero_shift,
            yhat, None, None, None, self.F_in, self.F_out, self.p.act_ids,
            self.p.wmax,
        )
        return yhat

    def anomaly(self, X: torch.Tensor, inv_err_scale: np.ndarray):
        """Fused forward + per-feature |err|*inv_scale + row L2 (y = scaled x)."""
        X = X.to(self.device, torch.float32).contiguous()
        B = X.shape[0]
        yhat = torch.empty(B, self.F_out, dtype=torch.float32, device=self.device)
        err = torch.empty(B, self.F_out, dtype=torch.float32, device=self.device)
        total = torch.empty(B, dtype=torch.float32, device=self.device)
        inv = torch.zeros(self.p.wmax, dtype=torch.float32, device=self.device)
        inv[: self.F_out] = torch.as_tensor(inv_err_scale, dtype=torch.float32, device=self.device)
        self.E.hg_forward(
            X, self.p.w_bf16, self.p.bias, self.identity_scale, self.zero_shift,
            yhat, err, total, inv, self.F_in, self.F_out, self.p.act_ids,
            self.p.wmax,
        )
        return yhat, err, total
