"""Training engine.

The reference delegates its hot loop to ``keras.Model.fit``;
this engine is the MI355X-native replacement. Two execution paths behind one
interface:

- **fused** (GPU, FeedForwardNet): the whole train step — minibatch gather,
  forward GEMM stack, MSE loss, backward, fp32-master Adam — runs as
  hand-written CDNA4 HIP kernels (gordo_amd.ops), one hipGraph per step.
- **eager** (CPU, or unsupported architectures): plain torch autograd loop.

Data-parallel training over RCCL/xGMI is layered on via
gordo_amd.parallel (flat-bucket all-reduce of the fused gradient buffer).
"""
from __future__ import annotations

import logging
import os
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np
import torch
from torch import nn

logger = logging.getLogger(__name__)


def resolve_device(device: str = "auto") -> torch.device:
    if device == "auto":
        # the fleet scheduler pins workers to GPUs via GORDO_DEVICE
        env = os.environ.get("GORDO_DEVICE")
        if env:
            return torch.device(env)
        return torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")
    return torch.device(device)


@dataclass
class TrainConfig:
    epochs: int = 10
    batch_size: int = 256
    lr: float = 1e-3
    betas: tuple = (0.9, 0.999)
    eps: float = 1e-8
    weight_decay: float = 0.0
    shuffle: bool = True
    validation_split: float = 0.0
    device: str = "auto"
    dtype: str = "bf16"  # compute dtype on GPU; params stay fp32
    loss: str = "mse"
    verbose: int = 0
    seed: Optional[int] = None
    use_graph: bool = True  # hipGraph-capture the fused train step

    @classmethod
    def from_kwargs(cls, kwargs: Dict) -> "TrainConfig":
        names = set(cls.__dataclass_fields__)
        return cls(**{k: v for k, v in kwargs.items() if k in names})


@dataclass
class TrainHistory:
    loss: List[float] = field(default_factory=list)
    val_loss: List[float] = field(default_factory=list)
    epochs: int = 0
    duration_sec: float = 0.0
    samples_per_sec: float = 0.0
    engine: str = "eager"

    def to_dict(self) -> Dict:
        return {
            "loss": self.loss,
            "val_loss": self.val_loss,
            "epochs": self.epochs,
            "duration_sec": self.duration_sec,
            "samples_per_sec": self.samples_per_sec,
            "engine": self.engine,
        }


class Trainer:
    """Train a torch module on in-memory arrays."""

    def __init__(self, config: TrainConfig):
        self.config = config
        self.device = resolve_device(config.device)

    # ------------------------------------------------------------------
    def fit(
        self,
        module: nn.Module,
        X: np.ndarray,
        y: np.ndarray,
        lookback: int = 0,
        lookahead: int = 0,
    ) -> TrainHistory:
        """Train module; X is 2-D [N, F]. For sequence models (lookback>0)
        windows are formed on the fly (CPU: strided views; GPU: gather)."""
        cfg = self.config
        if cfg.seed is not None:
            torch.manual_seed(cfg.seed)

        # Try the fused MI355X engine first (GPU + supported architecture)
        if self.device.type == "cuda":
            from ..ops import fused

            engine = fused.maybe_fused_trainer(module, self.device, cfg)
            if engine is not None:
                return engine.fit(X, y, lookback=lookback, lookahead=lookahead)
            logger.info(
                "Fused engine unavailable for %s; using eager torch path",
                type(module).__name__,
            )

        return self._fit_eager(module, X, y, lookback, lookahead)

    # ------------------------------------------------------------------
    def _fit_eager(
        self,
        module: nn.Module,
        X: np.ndarray,
        y: np.ndarray,
        lookback: int,
        lookahead: int,
    ) -> TrainHistory:
        cfg = self.config
        device = self.device
        module = module.to(device)
        Xt = torch.as_tensor(np.asarray(X, dtype=np.float32), device=device)
        yt = torch.as_tensor(np.asarray(y, dtype=np.float32), device=device)

        n_val = 0
        if cfg.validation_split > 0:
            n_total = len(Xt) - (lookback - 1 + lookahead if lookback else 0)
            n_val = int(n_total * cfg.validation_split)

        if lookback:
            n_windows = len(Xt) - lookback + 1 - lookahead
            if n_windows <= 0:
                raise ValueError(
                    f"Need > {lookback + lookahead} rows for lookback={lookback}"
                )
            idx_all = torch.arange(n_windows, device=device)
        else:
            idx_all = torch.arange(len(Xt), device=device)

        idx_train = idx_all[: len(idx_all) - n_val] if n_val else idx_all
        idx_val = idx_all[len(idx_all) - n_val :] if n_val else None

        opt = torch.optim.Adam(
            module.parameters(),
            lr=cfg.lr,
            betas=cfg.betas,
            eps=cfg.eps,
            weight_decay=cfg.weight_decay,
        )
        loss_fn = nn.MSELoss()
        use_bf16 = device.type == "cuda" and cfg.dtype == "bf16"

        def batch_of(idx: torch.Tensor):
            if lookback:
                # gather windows [b, lookback, F]
                w = idx.unsqueeze(1) + torch.arange(lookback, device=device).unsqueeze(0)
                xb = Xt[w]
                yb = yt[idx + lookback - 1 + lookahead]
            else:
                xb = Xt[idx]
                yb = yt[idx]
            return xb, yb

        history = TrainHistory(engine="eager")
        t0 = time.perf_counter()
        steps = 0
        for epoch in range(cfg.epochs):
            if cfg.shuffle:
                perm = torch.randperm(len(idx_train), device=device)
                epoch_idx = idx_train[perm]
            else:
                epoch_idx = idx_train
            total, count = 0.0, 0
            module.train()
            for start in range(0, len(epoch_idx), cfg.batch_size):
                idx = epoch_idx[start : start + cfg.batch_size]
                xb, yb = batch_of(idx)
                opt.zero_grad(set_to_none=True)
                if use_bf16:
                    with torch.autocast("cuda", dtype=torch.bfloat16):
                        out = module(xb)
                        loss = loss_fn(out.float(), yb)
                else:
                    loss = loss_fn(module(xb), yb)
                loss.backward()
                opt.step()
                total += loss.item() * len(idx)
                count += len(idx)
                steps += 1
            history.loss.append(total / max(count, 1))
            if idx_val is not None and len(idx_val):
                module.eval()
                with torch.no_grad():
                    xb, yb = batch_of(idx_val)
                    history.val_loss.append(loss_fn(module(xb), yb).item())
            if cfg.verbose:
                logger.info(
                    "epoch %d/%d loss=%.6f", epoch + 1, cfg.epochs, history.loss[-1]
                )
        if device.type == "cuda":
            torch.cuda.synchronize()
        history.epochs = cfg.epochs
        history.duration_sec = time.perf_counter() - t0
        n_samples = len(idx_train) * cfg.epochs
        history.samples_per_sec = n_samples / history.duration_sec if history.duration_sec else 0.0
        return history
