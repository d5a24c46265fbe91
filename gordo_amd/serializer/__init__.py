"""Serializer — the model-artifact (checkpoint) contract.

Parity: gordo_components/serializer/ — dict↔pipeline in both directions plus
directory dump/load. The directory written by ``dump`` IS the artifact
contract between builder and server (SURVEY.md §2.1 "Serializer"):

    <dir>/
      metadata.json       build + model metadata
      definition.yaml     into_definition(model) when derivable (inspectable)
      model.pkl           the picklable object graph (torch weights embedded
                          as bytes by TorchBaseEstimator.__getstate__)
      weights.safetensors flat tensor blobs for TorchBaseEstimator members —
                          the server's zero-pickle fast load path

The reference wrote one nested subdir per pipeline step with Keras HDF5 blobs
inside pickles; this format keeps the same information (structure in
definition.yaml, weights in safetensors, metadata.json at root) laid out for
fast mmap-load into 288 GB HBM.
"""
from __future__ import annotations

import io
import json
import logging
import os
import pickle
from contextlib import contextmanager
from typing import Any, Dict, Optional

import yaml

from ._from_definition import from_definition, pipeline_from_definition
from ._into_definition import into_definition, pipeline_into_definition

logger = logging.getLogger(__name__)

__all__ = [
    "from_definition",
    "into_definition",
    "pipeline_from_definition",
    "pipeline_into_definition",
    "dump",
    "dumps",
    "load",
    "loads",
    "load_metadata",
]


def _walk_estimators(obj: Any, prefix: str = ""):
    """Yield (prefix, TorchBaseEstimator) for every torch estimator in the
    object graph — the one traversal shared by dump (tensor export) and load
    (tensor restore), so safetensors key names always agree."""
    from sklearn.pipeline import FeatureUnion, Pipeline

    from ..model.models import TorchBaseEstimator

    if isinstance(obj, TorchBaseEstimator):
        yield prefix, obj
    elif isinstance(obj, Pipeline):
        for name, step in obj.steps:
            yield from _walk_estimators(step, f"{prefix}{name}.")
    elif isinstance(obj, FeatureUnion):
        for name, t in obj.transformer_list:
            yield from _walk_estimators(t, f"{prefix}{name}.")
    elif hasattr(obj, "base_estimator"):
        yield from _walk_estimators(obj.base_estimator, f"{prefix}base_estimator.")


def _collect_torch_tensors(obj: Any, prefix: str = "") -> Dict[str, "object"]:
    """name→tensor for safetensors export."""
    tensors: Dict[str, Any] = {}
    for pfx, est in _walk_estimators(obj, prefix):
        if est.model is not None:
            for k, v in est.model.state_dict().items():
                tensors[f"{pfx}model.{k}"] = v.detach().cpu().contiguous()
    return tensors


@contextmanager
def _externalized_weights():
    from ..model.models import TorchBaseEstimator

    TorchBaseEstimator._externalize_tls.on = True
    try:
        yield
    finally:
        TorchBaseEstimator._externalize_tls.on = False


def dump(obj: Any, dest_dir: str, metadata: Optional[Dict] = None):
    os.makedirs(dest_dir, exist_ok=True)
    # Tensors go to weights.safetensors (mmap-able flat blobs — THE weight
    # load path); the pickle then carries only the object graph + a marker
    # per estimator. If safetensors export fails, fall back to a
    # self-contained pickle with embedded weight bytes.
    externalized = False
    try:
        tensors = _collect_torch_tensors(obj)
        if tensors:
            from safetensors.torch import save_file

            save_file(tensors, os.path.join(dest_dir, "weights.safetensors"))
            externalized = True
    except Exception as e:
        logger.warning("weights.safetensors not written: %s", e)
    with open(os.path.join(dest_dir, "model.pkl"), "wb") as f:
        if externalized:
            with _externalized_weights():
                pickle.dump(obj, f)
        else:
            pickle.dump(obj, f)
    if metadata is not None:
        with open(os.path.join(dest_dir, "metadata.json"), "w") as f:
            json.dump(metadata, f, default=str, indent=1)
    try:
        definition = into_definition(obj)
        with open(os.path.join(dest_dir, "definition.yaml"), "w") as f:
            yaml.safe_dump(definition, f, default_flow_style=False)
    except Exception as e:
        logger.debug("definition.yaml not derivable for %s: %s", type(obj).__name__, e)


def dumps(obj: Any) -> bytes:
    buf = io.BytesIO()
    pickle.dump(obj, buf)
    return buf.getvalue()


def loads(data: bytes) -> Any:
    return pickle.loads(data)


def load(source_dir: str) -> Any:
    """Load an artifact dir: unpickle the object graph, then restore any
    externalized torch weights from weights.safetensors (zero torch-tensor
    bytes pass through the pickle on this path)."""
    path = os.path.join(source_dir, "model.pkl")
    if not os.path.exists(path):
        raise FileNotFoundError(f"No model.pkl under {source_dir}")
    with open(path, "rb") as f:
        obj = pickle.load(f)

    pending = [
        (pfx, est)
        for pfx, est in _walk_estimators(obj)
        if getattr(est, "_needs_external_weights", False)
    ]
    if pending:
        st_path = os.path.join(source_dir, "weights.safetensors")
        if not os.path.exists(st_path):
            raise FileNotFoundError(
                f"{source_dir}: model.pkl references external weights but "
                "weights.safetensors is missing"
            )
        from safetensors import safe_open

        with safe_open(st_path, framework="pt", device="cpu") as sf:
            keys = list(sf.keys())
            for pfx, est in pending:
                want = f"{pfx}model."
                sd = {k[len(want):]: sf.get_tensor(k) for k in keys if k.startswith(want)}
                if not sd:
                    raise KeyError(
                        f"weights.safetensors has no tensors under '{want}'"
                    )
                est.restore_weights(sd)
    return obj


def load_metadata(source_dir: str) -> Dict:
    path = os.path.join(source_dir, "metadata.json")
    if not os.path.exists(path):
        raise FileNotFoundError(f"No metadata.json under {source_dir}")
    with open(path) as f:
        return json.load(f)
