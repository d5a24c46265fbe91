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


def _collect_torch_tensors(obj: Any, prefix: str = "") -> Dict[str, "object"]:
    """Walk the object graph for TorchBaseEstimator members; return
    name→tensor for safetensors export."""
    from ..model.models import TorchBaseEstimator

    tensors: Dict[str, Any] = {}

    def walk(o: Any, pfx: str):
        from sklearn.pipeline import FeatureUnion, Pipeline

        if isinstance(o, TorchBaseEstimator) and o.model is not None:
            for k, v in o.model.state_dict().items():
                tensors[f"{pfx}model.{k}"] = v.detach().cpu().contiguous()
        elif isinstance(o, Pipeline):
            for name, step in o.steps:
                walk(step, f"{pfx}{name}.")
        elif isinstance(o, FeatureUnion):
            for name, t in o.transformer_list:
                walk(t, f"{pfx}{name}.")
        elif hasattr(o, "base_estimator"):
            walk(o.base_estimator, f"{pfx}base_estimator.")

    walk(obj, prefix)
    return tensors


def dump(obj: Any, dest_dir: str, metadata: Optional[Dict] = None):
    os.makedirs(dest_dir, exist_ok=True)
    with open(os.path.join(dest_dir, "model.pkl"), "wb") as f:
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
    try:
        tensors = _collect_torch_tensors(obj)
        if tensors:
            from safetensors.torch import save_file

            save_file(tensors, os.path.join(dest_dir, "weights.safetensors"))
    except Exception as e:
        logger.warning("weights.safetensors not written: %s", e)


def dumps(obj: Any) -> bytes:
    buf = io.BytesIO()
    pickle.dump(obj, buf)
    return buf.getvalue()


def loads(data: bytes) -> Any:
    return pickle.loads(data)


def load(source_dir: str) -> Any:
    path = os.path.join(source_dir, "model.pkl")
    if not os.path.exists(path):
        raise FileNotFoundError(f"No model.pkl under {source_dir}")
    with open(path, "rb") as f:
        return pickle.load(f)


def load_metadata(source_dir: str) -> Dict:
    path = os.path.join(source_dir, "metadata.json")
    if not os.path.exists(path):
        raise FileNotFoundError(f"No metadata.json under {source_dir}")
    with open(path) as f:
        return json.load(f)
