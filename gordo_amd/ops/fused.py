"""Fused MI355X training/inference engines.

``maybe_fused_trainer(module, device, cfg)`` returns a fused engine when the
architecture is supported on GPU (FeedForwardNet → fused hourglass kernels),
else None and the caller uses the eager path.
"""
from __future__ import annotations

import logging
from typing import Optional

import torch

from . import is_available

logger = logging.getLogger(__name__)


def maybe_fused_trainer(module, device: torch.device, cfg) -> Optional[object]:
    if device.type != "cuda" or not is_available():
        return None
    from ..model.torch_nets import FeedForwardNet, LSTMNet

    if isinstance(module, FeedForwardNet):
        from .hourglass_engine import FusedHourglassTrainer

        try:
            return FusedHourglassTrainer(module, device, cfg)
        except Exception as e:  # unsupported shape etc.
            logger.warning("Fused hourglass engine rejected module: %s", e)
            return None
    if isinstance(module, LSTMNet):
        from .lstm_engine import FusedLSTMTrainer

        try:
            return FusedLSTMTrainer(module, device, cfg)
        except Exception as e:
            logger.warning("Fused LSTM engine rejected module: %s", e)
        # widths beyond the HIP cell envelope: semi-fused GEMM-loop trainer
        # (vastly faster than torch.nn.LSTM's eager path on ROCm)
        from .wide_lstm import WideLSTMTrainer

        return WideLSTMTrainer(module, device, cfg)
    return None
