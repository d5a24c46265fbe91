"""Model contract.

Parity: gordo_components/model/base.py — ``GordoBase``: the extra surface
every gordo model provides on top of the sklearn estimator API.
"""
from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np


class GordoBase:
    def get_metadata(self) -> Dict[str, Any]:
        """Build-time metadata (history, params) merged into the model's
        metadata.json by the builder."""
        raise NotImplementedError()

    def score(
        self,
        X: np.ndarray,
        y: Optional[np.ndarray] = None,
        sample_weight: Optional[np.ndarray] = None,
    ) -> float:
        raise NotImplementedError()
