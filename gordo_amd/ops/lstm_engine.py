"""Fused LSTM training engine (Python driver for the CDNA4 persistent-cell
kernels in hip/lstm_kernels.hip — SURVEY.md §2.3 K4/K5/K6).

The reference trains Keras stacked LSTMs over rolling windows
(gordo_components/model/factories/lstm_autoencoder.py +
model/utils.py::create_keras_timeseriesgenerator). Here each training step is:

  per layer l:        lstm_fwd      — ONE kernel loops t=0..T-1 on-chip
                                      (layer 0 gathers windows straight from
                                      the resident dataset — no materialized
                                      [B,T,F] batch in HBM traffic terms)
  head fwd+bwd+loss:  hg_train_step — 1-layer dense head on h_{T-1}, MSE,
                                      dx_out = dL/dh_{T-1}
  per layer l (rev):  lstm_bwd_cell — reverse-time BPTT (dgates) with the
                      dx_seq = dgates x W_ih GEMM fused into the same pass,
                      then lstm_dw — split-K MFMA kernel producing
                      dWcat = dgates^T x [x | h_{t-1}] AND dbias from ONE
                      pass over dgates (rocBLAS is ~20x slower at this
                      single-tile, K=B*T shape)
  optimizer:          fp32-master Adam (torch foreach) + bf16 weight re-emit

Weight layout: per layer one packed [4W][2W] bf16 "wcat" at the LAYER's own
width tier W in {32, 64, 128, 256} (cols :W = W_ih zero-padded, W: = W_hh;
rows gate-major i,f,g,o each padded to W — same gate order as
torch.nn.LSTM) plus a fused fp32 bias b_ih+b_hh [4W].
Zero padding is gradient-invariant: padded h stays exactly 0 through the cell
(weights/bias 0 -> g=tanh(0)=0 -> c=0 -> h=0) and padded grads are 0.
"""
from __future__ import annotations

import logging
import math
import time
from typing import List, Optional

import numpy as np
import torch

from . import ext
from .hourglass_engine import ACT_IDS, WMAX, pad_rows_bf16
from ..runtime.trainer import TrainHistory

logger = logging.getLogger(__name__)

H4 = 4 * WMAX  # 256 (narrow variant; wide uses per-instance 4*W)
KCAT = 2 * WMAX  # 128
WMAX_WIDE = 128
WMAX_WIDE256 = 256
WMIN_NARROW = 32  # narrow cell tier (real widths <= 32; headline 25-wide cells)


def lstm_width(module) -> int:
    """WIDEST padded kernel tier across the stack (64/128/256 — the legacy
    whole-stack answer; since r2 each layer actually runs at its own tier,
    including the W=32 narrow one — see _LSTMPackedParams.Wl)."""
    dims = [module.n_features, module.n_features_out]
    for cell in module.cells:
        dims += [cell.hidden_size, cell.input_size]
    m = max(dims)
    if m <= WMAX:
        return WMAX
    return WMAX_WIDE if m <= WMAX_WIDE else WMAX_WIDE256


def lstm_supported(module) -> Optional[str]:
    """None if the fused path supports this LSTMNet, else the reason."""
    from ..model.torch_nets import LSTMNet

    if not isinstance(module, LSTMNet):
        return f"{type(module).__name__} is not LSTMNet"
    if module.n_features > WMAX_WIDE256:
        return f"n_features {module.n_features} > {WMAX_WIDE256}"
    if module.n_features_out > WMAX_WIDE256:
        return f"n_features_out {module.n_features_out} > {WMAX_WIDE256}"
    for cell in module.cells:
        if cell.hidden_size > WMAX_WIDE256 or cell.input_size > WMAX_WIDE256:
            return f"LSTM dims {cell.input_size}->{cell.hidden_size} > {WMAX_WIDE256}"
        if cell.num_layers != 1 or cell.bidirectional:
            return "only stacked unidirectional single-layer cells"
    if module.out_func not in ACT_IDS:
        return f"out_func {module.out_func!r} not fused"
    return None


def _kernels_for(E, W: int):
    """(fwd, bwd, dw) kernel triple for a width tier."""
    if W == WMIN_NARROW:
        return E.lstm_fwd_h32, E.lstm_bwd_h32, E.lstm_dw_h32
    if W == WMAX:
        return E.lstm_fwd, E.lstm_bwd_cell, E.lstm_dw
    if W == WMAX_WIDE:
        return E.lstm_fwd_wide, E.lstm_bwd_wide, E.lstm_dw_wide
    return E.lstm_fwd_wide256, E.lstm_bwd_wide256, E.lstm_dw_wide256


def _tier(width: int) -> int:
    """Smallest kernel width tier covering a real width."""
    if width <= WMIN_NARROW:
        return WMIN_NARROW
    if width <= WMAX:
        return WMAX
    return WMAX_WIDE if width <= WMAX_WIDE else WMAX_WIDE256


class _LSTMPackedParams:
    """Padded fp32 masters + bf16 kernel weights for the stacked LSTM + head.

    Each cell runs at its OWN width tier (64 register-resident / 128
    L2-streamed / 256 column-half-split): a 200-tag hourglass stack is
    [256, 256, 128, 256] instead of uniformly 256, and a 100-tag stack
    [128, 128, 64, 128] — the narrow tiers are several times cheaper per
    layer. Tier boundaries are bridged by zero-padded slice copies of the
    h/dh streams in the engine (~0.2 ms per boundary per step at B=16K,
    negligible vs the cell kernels)."""

    def __init__(self, module, device):
        self.module = module
        self.device = device
        self.Lc = len(module.cells)
        # per-cell tier: must cover both the (padded) input and the hidden
        self.Wl: List[int] = [
            _tier(max(cell.input_size, cell.hidden_size)) for cell in module.cells
        ]
        self.W = self.Wl[0]        # input padding width (dataset rows)
        self.H4 = 4 * self.W
        self.KCAT = 2 * self.W
        self.wcat_master: List[torch.Tensor] = []
        self.bias_master: List[torch.Tensor] = []
        self.wcat_bf16: List[torch.Tensor] = []
        self.wcat_t_bf16: List[torch.Tensor] = []
        self.h_real: List[int] = []
        for l, cell in enumerate(module.cells):
            H, I = cell.hidden_size, cell.input_size
            W = self.Wl[l]
            self.h_real.append(H)
            wcat = torch.zeros(4 * W, 2 * W, dtype=torch.float32, device=device)
            bias = torch.zeros(4 * W, dtype=torch.float32, device=device)
            w_ih = cell.weight_ih_l0.detach().to(device, torch.float32)  # [4H, I]
            w_hh = cell.weight_hh_l0.detach().to(device, torch.float32)  # [4H, H]
            b = (cell.bias_ih_l0 + cell.bias_hh_l0).detach().to(device, torch.float32)
            for g in range(4):
                wcat[g * W : g * W + H, :I] = w_ih[g * H : (g + 1) * H]
                wcat[g * W : g * W + H, W : W + H] = w_hh[g * H : (g + 1) * H]
                bias[g * W : g * W + H] = b[g * H : (g + 1) * H]
            self.wcat_master.append(wcat)
            self.bias_master.append(bias)
            self.wcat_bf16.append(wcat.to(torch.bfloat16).contiguous())
            self.wcat_t_bf16.append(wcat.t().contiguous().to(torch.bfloat16))
        # dense head on h_{T-1}: its own tier over (H_last, F_out) — floor
        # 64 (the hg head kernel's narrowest instantiation)
        W_head = max(WMAX, _tier(max(module.cells[-1].hidden_size, module.n_features_out)))
        self.W_head = W_head
        self.head_wm = torch.zeros(W_head, W_head, dtype=torch.float32, device=device)
        self.head_bm = torch.zeros(W_head, dtype=torch.float32, device=device)
        out_w, in_w = module.head.weight.shape
        self.head_wm[:out_w, :in_w] = module.head.weight.detach().to(device, torch.float32)
        self.head_bm[:out_w] = module.head.bias.detach().to(device, torch.float32)
        self.head_w = self.head_wm.to(torch.bfloat16).contiguous()
        self.head_wt = self.head_wm.t().contiguous().to(torch.bfloat16)
        self.head_bias = self.head_bm.clone()

    def refresh_bf16(self):
        for l in range(self.Lc):
            self.wcat_bf16[l].copy_(self.wcat_master[l].to(torch.bfloat16))
            self.wcat_t_bf16[l].copy_(self.wcat_master[l].t().to(torch.bfloat16))
        self.head_w.copy_(self.head_wm.to(torch.bfloat16))
        self.head_wt.copy_(self.head_wm.t().to(torch.bfloat16))
        self.head_bias.copy_(self.head_bm)

    def masters(self) -> List[torch.Tensor]:
        return [*self.wcat_master, *self.bias_master, self.head_wm, self.head_bm]

    def sync_to_module(self):
        with torch.no_grad():
            for l, cell in enumerate(self.module.cells):
                H, I = cell.hidden_size, cell.input_size
                W = self.Wl[l]
                wcat = self.wcat_master[l]
                bias = self.bias_master[l]
                for g in range(4):
                    cell.weight_ih_l0[g * H : (g + 1) * H].copy_(
                        wcat[g * W : g * W + H, :I].to(cell.weight_ih_l0.device)
                    )
                    cell.weight_hh_l0[g * H : (g + 1) * H].copy_(
                        wcat[g * W : g * W + H, W : W + H].to(cell.weight_hh_l0.device)
                    )
                    # split the fused bias evenly (torch keeps two bias vectors)
                    half = bias[g * W : g * W + H].to(cell.bias_ih_l0.device) * 0.5
                    cell.bias_ih_l0[g * H : (g + 1) * H].copy_(half)
                    cell.bias_hh_l0[g * H : (g + 1) * H].copy_(half)
            out_w, in_w = self.module.head.weight.shape
            self.module.head.weight.copy_(self.head_wm[:out_w, :in_w].to(self.module.head.weight.device))
            self.module.head.bias.copy_(self.head_bm[:out_w].to(self.module.head.bias.device))


class FusedLSTMTrainer:
    """One-GPU fused trainer for LSTMNet (KerasLSTM{AutoEncoder,Forecast} parity)."""

    def __init__(self, module, device: torch.device, cfg):
        reason = lstm_supported(module)
        if reason:
            raise ValueError(f"fused LSTM path unsupported: {reason}")
        self.module = module
        self.device = device
        self.cfg = cfg
        self.E = ext()
        if self.E is None:
            raise RuntimeError("HIP extension missing")

    # ------------------------------------------------------------------
    def fit(self, X: np.ndarray, y: np.ndarray, lookback: int = 144, lookahead: int = 0) -> TrainHistory:
        cfg = self.cfg
        device = self.device
        if cfg.seed is not None:
            torch.manual_seed(cfg.seed)
        T = int(lookback)
        E = self.E
        p = _LSTMPackedParams(self.module, device)
        self.p = p
        Lc = p.Lc
        F_out = self.module.n_features_out

        y_is_x = y is X or (X.shape == y.shape and np.shares_memory(X, y))
        Xd = pad_rows_bf16(np.asarray(X, np.float32), device, p.W)
        Yd = Xd if y_is_x else pad_rows_bf16(np.asarray(y, np.float32), device, p.W)
        N = Xd.shape[0]
        n_windows = N - T + 1 - lookahead
        if n_windows <= 0:
            raise ValueError(f"need > {T + lookahead} rows for lookback={T}")

        B = max(64, min(cfg.batch_size, n_windows))
        n_batches = max(1, math.ceil(n_windows / B))
        st = self._alloc_step_state(Xd, Yd, B, T, n_batches, n_windows, lookahead, F_out)

        opt = torch.optim.Adam(
            p.masters(), lr=cfg.lr, betas=cfg.betas, eps=cfg.eps, foreach=True
        )

        history = TrainHistory(engine="fused_hip_lstm")
        denom = float(n_batches) * B * F_out
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(cfg.epochs):
            self._fill_perm(st)
            for s in range(n_batches):
                self._step(st, s, opt)
            history.loss.append(float(st["loss_acc"].item()) / denom)
            st["loss_acc"].zero_()
        torch.cuda.synchronize()
        history.duration_sec = time.perf_counter() - t0
        history.epochs = cfg.epochs
        history.samples_per_sec = (
            cfg.epochs * n_batches * B / history.duration_sec if history.duration_sec else 0.0
        )
        p.sync_to_module()
        return history

    # ------------------------------------------------------------------
    @staticmethod
    def _dw_buffers(tier: int, device):
        """Per-tier dW split-K partial buffers (layers of one tier share)."""
        f32 = torch.float32
        if tier == WMIN_NARROW:
            # narrow split-K: 512 flat slabs like W=64
            return (
                torch.zeros(512, 4 * tier * 2 * tier, dtype=f32, device=device),
                torch.zeros(512, 4 * tier, dtype=f32, device=device),
            )
        if tier == WMAX:
            # split-K dW partials: 512 blocks = 2 blocks/CU (latency hiding)
            return (
                torch.zeros(512, 4 * tier * 2 * tier, dtype=f32, device=device),
                torch.zeros(512, 4 * tier, dtype=f32, device=device),
            )
        if tier == WMAX_WIDE:
            # wide: [4 quadrants][NB][256*128] + dbias [2][NB][256]
            return (
                torch.zeros(4, 256, 256 * 128, dtype=f32, device=device),
                torch.zeros(2, 256, 256, dtype=f32, device=device),
            )
        # W=256: [16 quadrants][NB][256*128] + dbias [4][NB][256]
        return (
            torch.zeros(16, 64, 256 * 128, dtype=f32, device=device),
            torch.zeros(4, 64, 256, dtype=f32, device=device),
        )

    def _alloc_step_state(self, Xd, Yd, B, T, n_batches, n_windows, lookahead, F_out):
        device = self.device
        p = self.p
        Lc = p.Lc
        bf16 = torch.bfloat16
        gates_dt = torch.uint8  # u8 fixed-point gate streams (common.h ga_q8_*)
        MT = int(getattr(self.E, "MT", 32))
        n_head_tiles = (B + MT - 1) // MT
        WH = p.W_head
        st = {
            "Xd": Xd, "Yd": Yd, "B": B, "T": T, "n_batches": n_batches,
            "n_windows": n_windows, "lookahead": lookahead, "F_out": F_out,
            "perm": torch.zeros(n_batches * B, dtype=torch.int32, device=device),
            "h_seq": [torch.zeros(B, T, p.Wl[l], dtype=bf16, device=device) for l in range(Lc)],
            "c_seq": [torch.zeros(B, T, p.Wl[l], dtype=bf16, device=device) for l in range(Lc)],
            # activated gates stored quantized u8 (common.h ga_q8_*: i,f,o
            # as v*255, g offset) — halves the biggest BPTT stream
            "gates_seq": [torch.zeros(B, T, 4 * p.Wl[l], dtype=gates_dt, device=device) for l in range(Lc)],
            "x_seq0": torch.zeros(B, T, p.Wl[0], dtype=bf16, device=device),
            "gw": [torch.zeros(4 * p.Wl[l] * 2 * p.Wl[l], dtype=torch.float32, device=device) for l in range(Lc)],
            "gb": [torch.zeros(4 * p.Wl[l], dtype=torch.float32, device=device) for l in range(Lc)],
            "h_last": torch.zeros(B, WH, dtype=bf16, device=device),
            "y_head": torch.zeros(B, WH, dtype=bf16, device=device),
            "dx_head": torch.zeros(B, WH, dtype=bf16, device=device),
            "counters0": torch.tensor([0, 1, 0], dtype=torch.int32, device=device),
            "head_dw": torch.zeros(n_head_tiles, 1, WH * WH, dtype=torch.bfloat16, device=device),
            "head_db": torch.zeros(n_head_tiles, 1, WH, dtype=torch.bfloat16, device=device),
            "loss_acc": torch.zeros(1, dtype=torch.float32, device=device),
            "gscale": 2.0 / (B * F_out),
            "act_out": [ACT_IDS[self.module.out_func]],
        }
        # tier-bridging staging + shared per-tier streams (mixed-width stacks)
        st["x_in"] = {
            l: torch.zeros(B, T, p.Wl[l], dtype=bf16, device=device)
            for l in range(1, Lc)
            if p.Wl[l] != p.Wl[l - 1]
        }
        st["dh_in"] = {
            l: torch.zeros(B, T, p.Wl[l], dtype=bf16, device=device)
            for l in range(Lc - 1)
            if p.Wl[l + 1] != p.Wl[l]
        }
        if WH != p.Wl[-1]:
            st["dh_head_in"] = torch.zeros(B, p.Wl[-1], dtype=bf16, device=device)
        bwd_tiers = sorted({p.Wl[l] for l in range(Lc)})
        dx_tiers = sorted({p.Wl[l] for l in range(1, Lc)})
        st["dgates"] = {
            t: torch.zeros(B, T, 4 * t, dtype=bf16, device=device) for t in bwd_tiers
        }
        # dh/dx stream per tier: a layer's fused-dx output, read as the next
        # (lower) layer's dh — aliased in place when both share the tier
        # (the kernel reads dh[gi][t] before writing dx[gi][t] per block)
        st["dhdx"] = {
            t: torch.zeros(B, T, t, dtype=bf16, device=device) for t in dx_tiers
        }
        st["dw_part"] = {}
        st["db_part"] = {}
        for t in bwd_tiers:
            st["dw_part"][t], st["db_part"][t] = self._dw_buffers(t, device)
        return st

    def _fill_perm(self, st):
        nw, need = st["n_windows"], st["perm"].numel()
        idx = torch.randperm(nw, device=self.device)
        if need > nw:
            idx = torch.cat([idx, torch.randint(0, nw, (need - nw,), device=self.device)])
        st["perm"].copy_(idx[:need].to(torch.int32))

    # ------------------------------------------------------------------
    def _forward(self, st, win: torch.Tensor):
        """Run the stacked fused forward for window starts ``win`` [B] int32.
        Each layer runs at its own width tier; tier boundaries are bridged by
        zero-padded slice copies of the h stream."""
        E, p = self.E, self.p
        B, T = st["B"], st["T"]
        for l in range(p.Lc):
            k_fwd, _, _ = _kernels_for(E, p.Wl[l])
            if l == 0:
                src, w = st["Xd"], win
            else:
                src = st["h_seq"][l - 1]
                if p.Wl[l] != p.Wl[l - 1]:
                    m = min(p.Wl[l], p.Wl[l - 1])
                    st["x_in"][l][:, :, :m].copy_(src[:, :, :m])
                    src = st["x_in"][l]
                w = None
            k_fwd(
                src, w, p.wcat_bf16[l], p.bias_master[l],
                st["h_seq"][l], st["gates_seq"][l], st["c_seq"][l],
                st["x_seq0"] if l == 0 else None, B, T,
            )
        mh = min(p.W_head, p.Wl[-1])
        st["h_last"][:, :mh].copy_(st["h_seq"][p.Lc - 1][:, T - 1, :mh])

    def _step(self, st, s: int, opt):
        win = st["perm"][s * st["B"] : (s + 1) * st["B"]]
        self._compute_grads(st, win)
        opt.step()
        opt.zero_grad(set_to_none=True)
        self.p.refresh_bf16()

    def _compute_grads(self, st, win: torch.Tensor):
        """Forward + head + BPTT; leaves fp32 grads on every master tensor."""
        E, p = self.E, self.p
        B, T = st["B"], st["T"]
        self._forward(st, win)

        # ---- head: dense fwd + MSE + bwd (dx_out = dL/dh_{T-1}) ----------
        tgt = win.long() + (T - 1 + st["lookahead"])
        if p.W_head == st["Yd"].shape[1]:
            torch.index_select(st["Yd"], 0, tgt, out=st["y_head"])
        else:
            m = min(p.W_head, st["Yd"].shape[1])
            st["y_head"][:, :m].copy_(st["Yd"].index_select(0, tgt)[:, :m])
        E.hg_train_step(
            st["h_last"], st["y_head"], st["counters0"],
            p.head_w, p.head_wt, p.head_bias, st["head_dw"], st["head_db"],
            st["loss_acc"], B, st["F_out"], st["act_out"], st["gscale"], False,
            st["dx_head"], p.W_head, 0, 0,
        )

        # ---- BPTT (fused dx) + split-K dW kernel, top layer down ---------
        # The per-tier dhdx stream doubles as the dx output: within a block,
        # dh_out[gi][t] is read (elementwise phase) before dx_seq[gi][t] is
        # written (GEMM phase of the same t), and blocks own disjoint gi
        # rows. Tier boundaries go through zero-padded dh_in staging copies.
        for l in reversed(range(p.Lc)):
            Wt = p.Wl[l]
            _, k_bwd, k_dw = _kernels_for(E, Wt)
            last_only = l == p.Lc - 1
            if last_only:
                if p.W_head == Wt:
                    dh = st["dx_head"]
                else:
                    m = min(p.W_head, Wt)
                    st["dh_head_in"][:, :m].copy_(st["dx_head"][:, :m])
                    dh = st["dh_head_in"]
            else:
                up = p.Wl[l + 1]
                if up == Wt:
                    dh = st["dhdx"][Wt]
                else:
                    m = min(up, Wt)
                    st["dh_in"][l][:, :, :m].copy_(st["dhdx"][up][:, :, :m])
                    dh = st["dh_in"][l]
            dx_out = st["dhdx"][Wt] if l > 0 else None
            k_bwd(
                dh, st["gates_seq"][l], st["c_seq"][l], p.wcat_t_bf16[l],
                st["dgates"][Wt], dx_out, B, T, last_only,
            )
            if l == 0:
                x_seq = st["x_seq0"]
            elif p.Wl[l] != p.Wl[l - 1]:
                x_seq = st["x_in"][l]
            else:
                x_seq = st["h_seq"][l - 1]
            k_dw(
                st["dgates"][Wt], x_seq, st["h_seq"][l],
                st["dw_part"][Wt], st["db_part"][Wt], B * T, T,
            )
            if Wt in (WMIN_NARROW, WMAX):
                torch.sum(st["dw_part"][Wt], dim=0, out=st["gw"][l])
                torch.sum(st["db_part"][Wt], dim=0, out=st["gb"][l])
                p.wcat_master[l].grad = st["gw"][l].view(4 * Wt, 2 * Wt)
                p.bias_master[l].grad = st["gb"][l]
            else:
                # quadrant tiles [gh][ih] of [256][128] -> gw [4W][2W]
                ng = 4 * Wt // 256
                ni = 2 * Wt // 128
                q = st["dw_part"][Wt].sum(1).view(ng, ni, 256, 128)
                gw = st["gw"][l].view(4 * Wt, 2 * Wt)
                for gh in range(ng):
                    for ih in range(ni):
                        gw[gh * 256 : (gh + 1) * 256, ih * 128 : (ih + 1) * 128].copy_(q[gh, ih])
                st["gb"][l].view(ng, 256).copy_(st["db_part"][Wt].sum(1))
                p.wcat_master[l].grad = gw
                p.bias_master[l].grad = st["gb"][l]

        # ---- head grads from the hg slabs (slab layout is [in][out]) -----
        p.head_wm.grad = (
            st["head_dw"].sum(0, dtype=torch.float32).view(p.W_head, p.W_head).t().contiguous()
        )
        p.head_bm.grad = st["head_db"].sum(0, dtype=torch.float32).view(p.W_head)


class LSTMStepTrainer:
    """Step-oriented fused LSTM trainer (bench + fleet building; DP-capable).

    DP exchanges ONE flat fp32 grad bucket per step over RCCL (the whole
    stacked model is ~0.5 MB — latency-bound on xGMI, so a single fused
    all-reduce, mirroring parallel/ddp.py's hourglass trainer).
    """

    def __init__(self, module, X: np.ndarray, batch_size: int, cfg, dp=None, lookahead: int = 0):
        from ..parallel.ddp import DPContext

        self.dp = dp or DPContext()
        self.device = self.dp.device
        self.cfg = cfg
        self.module = module
        self.tr = FusedLSTMTrainer(module, self.device, cfg)
        p = _LSTMPackedParams(module, self.device)
        self.tr.p = p
        self.p = p
        # ranks start from identical weights
        if self.dp.enabled:
            for t in p.masters():
                self.dp.broadcast(t)
            p.refresh_bf16()
        T = module.lookback_window
        Xd = pad_rows_bf16(np.asarray(X, np.float32), self.device, p.W)
        n_windows = Xd.shape[0] - T + 1 - lookahead
        B = max(64, min(batch_size, n_windows))
        self.B = B
        n_batches = max(1, math.ceil(n_windows / B))
        self.st = self.tr._alloc_step_state(
            Xd, Xd, B, T, n_batches, n_windows, lookahead, module.n_features_out
        )
        self.tr._fill_perm(self.st)
        self.opt = torch.optim.Adam(
            p.masters(), lr=cfg.lr, betas=cfg.betas, eps=cfg.eps, foreach=True
        )
        self._s = 0
        if self.dp.enabled:
            self._flat = torch.zeros(
                sum(t.numel() for t in p.masters()), dtype=torch.float32, device=self.device
            )

    def steps(self, n: int):
        # ms-scale steps: per-step Python overhead is negligible here (the
        # hourglass trainer's pair-graph trick buys nothing at 10 ms/step)
        for _ in range(n):
            self.step()

    def step(self):
        st = self.st
        win = st["perm"][self._s * self.B : (self._s + 1) * self.B]
        self.tr._compute_grads(st, win)
        if self.dp.enabled:
            ms = self.p.masters()
            off = 0
            for t in ms:
                n = t.numel()
                self._flat[off : off + n].copy_(t.grad.view(-1))
                off += n
            self.dp.all_reduce_mean(self._flat)
            off = 0
            for t in ms:
                n = t.numel()
                t.grad.view(-1).copy_(self._flat[off : off + n])
                off += n
        self.opt.step()
        self.opt.zero_grad(set_to_none=True)
        self.p.refresh_bf16()
        self._s += 1
        if self._s >= st["n_batches"]:
            self.tr._fill_perm(st)
            self._s = 0

    def synchronize(self):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)

    # mid-training checkpoint: masters + Adam state + batch
    # cursor, so a restarted DP job resumes the epoch where it stopped
    def state_dict(self) -> dict:
        return {
            "masters": [t.detach().cpu() for t in self.p.masters()],
            "opt": self.opt.state_dict(),
            "s": self._s,
        }

    def load_state_dict(self, sd: dict):
        for dst, src in zip(self.p.masters(), sd["masters"]):
            with torch.no_grad():
                dst.copy_(src.to(self.device))
        self.opt.load_state_dict(sd["opt"])
        self._s = int(sd["s"])
        self.p.refresh_bf16()

    def save_checkpoint(self, path: str):
        torch.save(self.state_dict(), path)

    def load_checkpoint(self, path: str):
        self.load_state_dict(torch.load(path, weights_only=False))


class FusedLSTMForward:
    """Serving/predict forward through the fused cell kernels."""

    def __init__(self, module, device: torch.device):
        reason = lstm_supported(module)
        if reason:
            raise ValueError(f"fused LSTM path unsupported: {reason}")
        self.E = ext()
        self.device = device
        self.module = module
        self.p = _LSTMPackedParams(module, device)

    def predict(self, X: np.ndarray, lookahead: int = 0, batch_size: int = 16384) -> np.ndarray:
        """X [N, F] -> predictions [N - T + 1 - lookahead, F_out].

        Windows whose target row (window end + lookahead) falls beyond X are
        dropped so downstream trailing-offset alignment (diff.py,
        metric_wrapper, make_base_dataframe) pairs prediction i with
        y[i + T - 1 + lookahead] — matching rolling_windows' training targets
        and the reference's forecast predict row count.
        """
        p = self.p
        T = self.module.lookback_window
        F_out = self.module.n_features_out
        Xd = pad_rows_bf16(np.asarray(X, np.float32), self.device, p.W)
        N = Xd.shape[0]
        n_windows = N - T + 1 - lookahead
        if n_windows <= 0:
            raise ValueError(f"need >= {T + lookahead} rows for lookback={T}")
        outs = []
        bf16 = torch.bfloat16
        B = min(batch_size, n_windows)
        h_seq = [torch.zeros(B, T, p.Wl[l], dtype=bf16, device=self.device) for l in range(p.Lc)]
        c_seq = {t: torch.zeros(B, T, t, dtype=bf16, device=self.device) for t in set(p.Wl)}
        gates = {t: torch.zeros(B, T, 4 * t, dtype=torch.uint8, device=self.device) for t in set(p.Wl)}
        x_in = {
            l: torch.zeros(B, T, p.Wl[l], dtype=bf16, device=self.device)
            for l in range(1, p.Lc)
            if p.Wl[l] != p.Wl[l - 1]
        }
        for s in range(0, n_windows, B):
            n = min(B, n_windows - s)
            win = torch.arange(s, s + n, dtype=torch.int32, device=self.device)
            if n < B:
                win = torch.cat([win, torch.zeros(B - n, dtype=torch.int32, device=self.device)])
            for l in range(p.Lc):
                k_fwd, _, _ = _kernels_for(self.E, p.Wl[l])
                if l == 0:
                    src, w = Xd, win
                else:
                    src = h_seq[l - 1]
                    if p.Wl[l] != p.Wl[l - 1]:
                        m = min(p.Wl[l], p.Wl[l - 1])
                        x_in[l][:, :, :m].copy_(src[:, :, :m])
                        src = x_in[l]
                    w = None
                k_fwd(
                    src, w, p.wcat_bf16[l], p.bias_master[l],
                    h_seq[l], gates[p.Wl[l]], c_seq[p.Wl[l]], None, B, T,
                )
            H_last = self.module.cells[-1].hidden_size
            h_last = h_seq[p.Lc - 1][:, T - 1, :H_last].float()
            yhat = torch.addmm(
                p.head_bm[:F_out], h_last, p.head_wm[:F_out, :H_last].t()
            )[:n]
            af = self.module.out_func
            if af and af != "linear":
                yhat = {"tanh": torch.tanh, "relu": torch.relu, "sigmoid": torch.sigmoid}[af](yhat)
            outs.append(yhat.cpu().numpy())
        return np.concatenate(outs, axis=0)
