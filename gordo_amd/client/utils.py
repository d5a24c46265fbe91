"""Client-side value types (parity: gordo_components/client/utils.py)."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

import pandas as pd


@dataclass
class EndpointMetadata:
    target_name: str
    endpoint: str
    healthy: bool = False
    metadata: Dict[str, Any] = field(default_factory=dict)

    @property
    def tag_list(self) -> List[str]:
        ds = self.metadata.get("metadata", self.metadata).get("dataset", {})
        return [t["name"] for t in ds.get("tag_list", [])]

    @property
    def resolution(self) -> Optional[str]:
        ds = self.metadata.get("metadata", self.metadata).get("dataset", {})
        return ds.get("resolution")

    @property
    def model_offset(self) -> int:
        md = self.metadata.get("metadata", self.metadata).get("model", {})
        return int(md.get("model-offset", 0) or 0)


@dataclass
class PredictionResult:
    name: str
    predictions: Optional[pd.DataFrame]
    error_messages: List[str] = field(default_factory=list)


def make_date_ranges(
    start: pd.Timestamp, end: pd.Timestamp, max_request_size_td: pd.Timedelta
) -> List[Tuple[pd.Timestamp, pd.Timestamp]]:
    """Split [start, end) into chunks of at most max_request_size_td."""
    out = []
    cur = start
    while cur < end:
        nxt = min(cur + max_request_size_td, end)
        out.append((cur, nxt))
        cur = nxt
    return out
