"""MI355X-native compute ops.

Hand-written CDNA4 HIP kernels (gfx950) for the hot paths identified in
SURVEY.md §2.3 (K1–K10): fused hourglass forward/backward, MSE loss+grad,
rolling-window gather, min-max scaling, feature-wise anomaly score, fused
Adam, and the persistent 4-gate LSTM cell.

Policy (per the build rules): on a GPU box the HIP extension MUST load — a
missing extension raises instead of silently falling back to eager torch.
On CPU-only machines the extension is not required (eager path covers CPU
tests).
"""
from __future__ import annotations

import logging

import torch

logger = logging.getLogger(__name__)

_EXT = None
_EXT_ERR: Exception | None = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import gordo_amd_hip  # built in-tree by setup.py build_ext --inplace

        _EXT = gordo_amd_hip
    except ImportError as e:
        _EXT_ERR = e
        if torch.cuda.is_available():
            raise RuntimeError(
                "gordo_amd_hip extension is not built but a GPU is present. "
                "Run `python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
                f"Import error: {e}"
            ) from e
        logger.debug("HIP extension unavailable (CPU-only host): %s", e)
    return _EXT


def ext():
    """The native extension module, or raise if on GPU and missing."""
    return _load_extension()


def is_available() -> bool:
    if not torch.cuda.is_available():
        return False
    return _load_extension() is not None
