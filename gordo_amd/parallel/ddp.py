"""Data-parallel training over RCCL/xGMI.

The reference has no distributed compute (fleet parallelism only, SURVEY.md
§2.4); this is the MI355X-native addition: one process per GPU, gradients
exchanged as ONE flat fp32 bucket per step (the hourglass model is KB-scale,
so the all-reduce is latency-bound — a single fused bucket is the right shape
for xGMI, not many small ones).

Step pipeline per rank (fused GPU path):
    hg_train_step (gather+fwd+loss+bwd, per-block bf16 slabs)
    hg_flat_reduce (slabs -> flat bucket + ping-pong counter bump)
    all_reduce(bucket, AVG) over RCCL        <- the only comm
    hg_adam_flat (reads the bumped counter slot)
Single GPU: hg_train_step + hg_reduce_adam (the whole tail in one kernel);
both variants replay from per-parity hipGraphs plus 2/4/8/16-step
multi-step graphs (widest tier parity allows per replay).

CPU path (gloo, used by the world_size=2 CPU tests here): eager torch autograd
with the same single-bucket all-reduce semantics.
"""
from __future__ import annotations

import logging
import math
import os
from typing import Optional

import numpy as np
import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)


class DPContext:
    """Thin wrapper around torch.distributed process-group state."""

    def __init__(self, backend: Optional[str] = None, device: Optional[torch.device] = None):
        self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
        self.rank = int(os.environ.get("RANK", "0"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", str(self.rank)))
        if device is None:
            if torch.cuda.is_available():
                device = torch.device(f"cuda:{self.local_rank}")
            else:
                device = torch.device("cpu")
        self.device = device
        self.enabled = self.world_size > 1
        if self.enabled and not dist.is_initialized():
            if backend is None:
                backend = "nccl" if device.type == "cuda" else "gloo"
            if device.type == "cuda":
                torch.cuda.set_device(device)
            # watchdog: a hung rank aborts the job instead of deadlocking it
            import datetime

            timeout_s = float(os.environ.get("GORDO_DIST_TIMEOUT", "600"))
            dist.init_process_group(
                backend=backend, timeout=datetime.timedelta(seconds=timeout_s)
            )
        self.backend = backend

    def all_reduce_mean(self, t: torch.Tensor):
        if self.enabled:
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
            t.div_(self.world_size)

    def barrier(self):
        if self.enabled:
            # device_ids is only meaningful (and only accepted) for nccl
            if self.device.type == "cuda" and self.backend == "nccl":
                dist.barrier(device_ids=[self.device.index])
            else:
                dist.barrier()

    def broadcast(self, t: torch.Tensor, src: int = 0):
        if self.enabled:
            dist.broadcast(t, src=src)


class HourglassStepTrainer:
    """Step-oriented trainer for the hourglass AE (bench + fleet building).

    Exposes ``step()`` so callers own the loop (warmup/timed regions); weights
    live on device; ``sync_to_module()`` copies them back.
    """

    def __init__(self, module, X: np.ndarray, batch_size: int, cfg, dp: Optional[DPContext] = None):
        self.dp = dp or DPContext()
        self.device = self.dp.device
        self.cfg = cfg
        self.module = module
        self.B = batch_size
        self.X = X

        self.fused = False
        if self.device.type == "cuda":
            from ..ops import is_available
            from ..ops.hourglass_engine import supported

            if is_available() and supported(module) is None:
                self.fused = True
        if self.fused:
            self._init_fused()
        else:
            self._init_eager()

    # ------------------------------------------------------------------
    def _init_fused(self):
        from ..ops import ext
        from ..ops.hourglass_engine import _PackedParams, pad_rows_bf16

        self.E = ext()
        device = self.device
        p = _PackedParams(self.module, device)
        WM = p.wmax
        # ranks must start from identical weights
        self.dp.broadcast(p.master)
        p.bias.copy_(p.master[p.PW:])
        w = p.master[: p.PW].view(p.L, WM, WM)
        p.w_bf16.copy_(w.to(torch.bfloat16))
        p.wt_bf16.copy_(w.transpose(1, 2).to(torch.bfloat16))
        self.p = p
        self.F_out = p.widths[-1]
        self.Xd = pad_rows_bf16(np.asarray(self.X, np.float32), device, WM)
        N = self.Xd.shape[0]
        self.N = N
        # GORDO_W1=1 opts into the wave-local W=64 step — measured slower
        # at the headline batch (see hourglass_engine.py / docs/ROADMAP.md)
        self._use_w1 = (
            WM == 64
            and hasattr(self.E, "hg_train_step_w1")
            and os.environ.get("GORDO_W1") == "1"
        )
        self._train_step = self.E.hg_train_step_w1 if self._use_w1 else self.E.hg_train_step
        MT = int(getattr(self.E, "MT_W1", 16)) if self._use_w1 else int(getattr(self.E, "MT", 32))
        self.n_tiles = (self.B + MT - 1) // MT
        self.dw = torch.zeros(self.n_tiles, p.L, WM * WM, dtype=torch.bfloat16, device=device)
        self.db = torch.zeros(self.n_tiles, p.L, WM, dtype=torch.bfloat16, device=device)
        self.gflat = torch.zeros(p.P, dtype=torch.float32, device=device)
        self.loss_acc = torch.zeros(1, dtype=torch.float32, device=device)
        # counters [batch idx, adam t, epoch]: PING-PONG pair — step q reads
        # slot q&1 and writes the bumped values to the other slot (bumping
        # inside hg_reduce_adam/hg_flat_reduce would race their own readers
        # on a single tensor). self.counters IS slot 0 (checkpoint name).
        self.counters = torch.tensor([0, 0, 0], dtype=torch.int32, device=device)
        self.counters_b = torch.tensor([0, 0, 0], dtype=torch.int32, device=device)
        self._cnt = [self.counters, self.counters_b]
        self._parity = 0
        self.G = min(16, self.n_tiles)
        self.partial = torch.zeros(self.G, p.P, dtype=torch.float32, device=device)
        self.m = torch.zeros_like(p.master)
        self.v = torch.zeros_like(p.master)
        self.n_batches = max(1, math.ceil(N / self.B))
        from ..ops.hourglass_engine import feistel_kbits

        self.kbits = feistel_kbits(N)
        # every rank shuffles its own shard with its own seed
        self.seed = 0x5EED ^ (self.dp.rank * 0x9E3779B1)
        self.gscale = 2.0 / (self.B * self.F_out)
        self._graphs = None
        self._multi_graphs = None
        # GORDO_NO_GRAPH=1 is the operational escape hatch for the captured
        # step (e.g. if an RCCL build misbehaves inside graph replay)
        if getattr(self.cfg, "use_graph", True) and not os.environ.get("GORDO_NO_GRAPH"):
            self._try_capture_graph()

    def _try_capture_graph(self):
        """Capture the whole step (train+reduce+all_reduce+adam+bump) into a
        hipGraph; counters/perm live in device memory so replays iterate.
        Single GPU captures one graph per counter parity (the fused
        hg_reduce_adam ping-pongs the counter slots)."""
        state = (
            self.p.master.clone(), self.m.clone(), self.v.clone(),
            self.counters.clone(),
        )
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for q in range(2):  # NCCL/RCCL warmup; both parities
                    self._fused_step_raw(q)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            graphs = []
            for q in range(2):  # one graph per counter parity
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self._fused_step_raw(q)
                graphs.append(g)
            # even-parity multi-step graphs: 2/4/8/16 whole steps per replay
            # cut the per-step Python/launch overhead (visible once the
            # kernels shrank to ~30 us/step; 4->8 steps measured +2.9% on
            # the headline bench)
            self._multi_graphs = []
            for size in (2, 4, 8, 16):
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    for i in range(size):
                        self._fused_step_raw(i & 1)
                self._multi_graphs.append((size, g))
            self._multi_graphs.reverse()  # widest first for the greedy replay
            # restore pre-warmup state
            self.p.master.copy_(state[0]); self.m.copy_(state[1]); self.v.copy_(state[2])
            self.counters.copy_(state[3])
            self._parity = 0
            self.p.bias.copy_(self.p.master[self.p.PW:])
            WM = self.p.wmax
            w = self.p.master[: self.p.PW].view(self.p.L, WM, WM)
            self.p.w_bf16.copy_(w.to(torch.bfloat16))
            self.p.wt_bf16.copy_(w.transpose(1, 2).contiguous().to(torch.bfloat16))
            self.loss_acc.zero_()
            torch.cuda.synchronize()
            self._graphs = graphs
            logger.info("train step captured in hipGraph (world=%d)", self.dp.world_size)
        except Exception as e:
            logger.warning("hipGraph capture failed (%s); plain launches", e)
            self._graphs = None
            self._multi_graphs = None
            # warm-up may have stepped the real state before the failure:
            # restore so the raw-launch run still starts pristine
            torch.cuda.synchronize()
            self.p.master.copy_(state[0]); self.m.copy_(state[1]); self.v.copy_(state[2])
            self.counters.copy_(state[3])
            self._parity = 0
            self.p.bias.copy_(self.p.master[self.p.PW:])
            WM = self.p.wmax
            w = self.p.master[: self.p.PW].view(self.p.L, WM, WM)
            self.p.w_bf16.copy_(w.to(torch.bfloat16))
            self.p.wt_bf16.copy_(w.transpose(1, 2).contiguous().to(torch.bfloat16))
            self.loss_acc.zero_()

    def _fused_step(self):
        if self._graphs is not None:
            self._graphs[self._parity].replay()
        else:
            self._fused_step_raw(self._parity)
        self._parity ^= 1

    def steps(self, n: int):
        """Run n whole training steps, using the widest multi-step graph
        (8/4/2 steps) parity allows — same work as n step() calls, fewer
        replays."""
        if not self.fused or self._graphs is None:
            for _ in range(n):
                self.step()
            return
        i = 0
        tiers = getattr(self, "_multi_graphs", None) or []
        while i < n:
            if self._parity == 0:
                for size, g in tiers:  # widest first; parity 0 -> 0
                    if n - i >= size:
                        g.replay()
                        i += size
                        break
                else:
                    self._fused_step()
                    i += 1
            else:
                self._fused_step()
                i += 1

    def _fused_step_raw(self, q: int = 0):
        p = self.p
        cfg = self.cfg
        WM = p.wmax
        cr, cw = self._cnt[q], self._cnt[q ^ 1]
        self._train_step(
            self.Xd, self.Xd, cr, p.w_bf16, p.wt_bf16, p.bias,
            self.dw, self.db, self.loss_acc, self.B, self.F_out, p.act_ids,
            self.gscale, True, None, WM, self.kbits, self.seed,
            [int(x) for x in p.widths],
        )
        if self.dp.enabled:
            # fused slabs -> flat bucket (+ ping-pong counter bump); the
            # Adam kernel runs after the collective and reads the bumped slot
            self.E.hg_flat_reduce(
                self.dw, self.db, self.gflat, cr, cw, self.n_tiles, p.L,
                self.n_batches, WM,
            )
            self.dp.all_reduce_mean(self.gflat)
            self.E.hg_adam_flat(
                p.master, self.m, self.v, self.gflat, p.w_bf16, p.wt_bf16, p.bias,
                cw, p.L, cfg.lr, cfg.betas[0], cfg.betas[1], cfg.eps, WM,
            )
        else:
            # fused slab reduce + Adam with ping-pong counters
            self.E.hg_reduce_adam(
                self.dw, self.db, p.master, self.m, self.v, p.w_bf16,
                p.wt_bf16, p.bias, cr, cw, self.n_tiles, p.L,
                self.n_batches, cfg.lr, cfg.betas[0], cfg.betas[1], cfg.eps, WM,
            )

    # ------------------------------------------------------------------
    def _init_eager(self):
        device = self.device
        self.module = self.module.to(device)
        self.dp_params = list(self.module.parameters())
        if self.dp.enabled:
            flat = torch.nn.utils.parameters_to_vector(self.dp_params)
            self.dp.broadcast(flat)
            torch.nn.utils.vector_to_parameters(flat, self.dp_params)
        self.Xt = torch.as_tensor(np.asarray(self.X, np.float32), device=device)
        self.opt = torch.optim.Adam(
            self.dp_params, lr=self.cfg.lr, betas=self.cfg.betas, eps=self.cfg.eps
        )
        self.loss_fn = torch.nn.MSELoss()
        self._gen = torch.Generator(device="cpu")
        self._gen.manual_seed(1234 + self.dp.rank)

    def _next_indices(self) -> torch.Tensor:
        """Minibatch row indices for the eager path; overridable so DP
        semantics tests can pin deterministic batches."""
        return torch.randint(0, len(self.Xt), (self.B,), generator=self._gen)

    def _eager_step(self):
        idx = self._next_indices().to(self.device)
        xb = self.Xt[idx]
        self.opt.zero_grad(set_to_none=True)
        use_bf16 = self.device.type == "cuda" and self.cfg.dtype == "bf16"
        if use_bf16:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                out = self.module(xb)
                loss = self.loss_fn(out.float(), xb)
        else:
            loss = self.loss_fn(self.module(xb), xb)
        loss.backward()
        if self.dp.enabled:
            grads = torch.nn.utils.parameters_to_vector(
                [p.grad for p in self.dp_params]
            )
            self.dp.all_reduce_mean(grads)
            offset = 0
            for par in self.dp_params:
                n = par.grad.numel()
                par.grad.copy_(grads[offset : offset + n].view_as(par.grad))
                offset += n
        self.opt.step()
        self._last_loss = loss

    # ------------------------------------------------------------------
    def step(self):
        if self.fused:
            self._fused_step()
        else:
            self._eager_step()

    def synchronize(self):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)

    def epoch_loss(self) -> float:
        if self.fused:
            val = float(self.loss_acc.item())
            self.loss_acc.zero_()
            return val
        return float(self._last_loss.item()) if hasattr(self, "_last_loss") else float("nan")

    def sync_to_module(self):
        if self.fused:
            self.p.sync_to_module()

    def params_vector(self) -> torch.Tensor:
        if self.fused:
            return self.p.master.detach().clone()
        return torch.nn.utils.parameters_to_vector(self.dp_params).detach().clone()

    # ------------------------------------------------------------------
    # Checkpoint/resume: the serializer directory is the
    # whole-model checkpoint; these cover mid-training state for long DP
    # runs — full optimizer state, so a restarted job continues bit-exact.
    def state_dict(self) -> dict:
        if self.fused:
            return {
                "fused": True,
                "master": self.p.master.detach().cpu(),
                "m": self.m.detach().cpu(),
                "v": self.v.detach().cpu(),
                # the LIVE slot: the one the next step will read (DP keeps
                # everything in slot 0; single-GPU ping-pongs the slots)
                "counters": self._cnt[self._parity].detach().cpu()
                if not self.dp.enabled
                else self.counters.detach().cpu(),
            }
        return {
            "fused": False,
            "params": torch.nn.utils.parameters_to_vector(self.dp_params).detach().cpu(),
            "opt": self.opt.state_dict(),
            "gen": self._gen.get_state(),
        }

    def load_state_dict(self, sd: dict):
        if self.fused != bool(sd.get("fused")):
            raise ValueError("checkpoint engine mismatch (fused vs eager)")
        if self.fused:
            p = self.p
            p.master.copy_(sd["master"].to(self.device))
            self.m.copy_(sd["m"].to(self.device))
            self.v.copy_(sd["v"].to(self.device))
            # restore into slot 0 and reset parity: bit-exact regardless of
            # the parity the checkpoint was taken at
            self.counters.copy_(sd["counters"].to(self.device))
            self._parity = 0
            p.bias.copy_(p.master[p.PW:])
            WM = p.wmax
            w = p.master[: p.PW].view(p.L, WM, WM)
            p.w_bf16.copy_(w.to(torch.bfloat16))
            p.wt_bf16.copy_(w.transpose(1, 2).contiguous().to(torch.bfloat16))
        else:
            torch.nn.utils.vector_to_parameters(
                sd["params"].to(self.device), self.dp_params
            )
            self.opt.load_state_dict(sd["opt"])
            if "gen" in sd:
                self._gen.set_state(sd["gen"])

    def save_checkpoint(self, path: str):
        torch.save(self.state_dict(), path)

    def load_checkpoint(self, path: str):
        self.load_state_dict(torch.load(path, weights_only=False))
