"""ModelBuilder — the train-time runtime.

Parity: gordo_components/builder/build_model.py — orchestrates:
dataset.from_dict → get_data() → from_definition(model_config) →
cross-validation (TimeSeriesSplit, per-tag + aggregate metrics via
metric_wrapper) → final fit on the full range (unless cv_mode=cross_val_only)
→ metadata assembly → serializer.dump → md5(config) registration in the disk
registry for cache reuse.
"""
from __future__ import annotations

import hashlib
import json
import logging
import os
import shutil
import time
from typing import Any, Dict, Optional, Tuple, Union

import numpy as np
import pandas as pd
from sklearn.metrics import (
    explained_variance_score,
    mean_absolute_error,
    mean_squared_error,
    r2_score,
)
from sklearn.model_selection import TimeSeriesSplit

from .. import __version__
from .cv import run_time_series_cv
from ..dataset import dataset_from_config
from ..machine import Machine
from ..model.anomaly.base import AnomalyDetectorBase
from ..model.base import GordoBase
from ..serializer import dump, from_definition
from ..util import disk_registry

logger = logging.getLogger(__name__)


class _phase_range:
    """roctx range around a build phase (torch.cuda.nvtx -> roctx on ROCm);
    makes builder phases visible in rocprof timelines. No-op off-GPU."""

    def __init__(self, name: str):
        self.name = name

    def __enter__(self):
        try:
            import torch

            if torch.cuda.is_available():
                torch.cuda.nvtx.range_push(self.name)
                self._active = True
                return self
        except Exception:
            pass
        self._active = False
        return self

    def __exit__(self, *exc):
        if getattr(self, "_active", False):
            import torch

            torch.cuda.nvtx.range_pop()
        return False

METRIC_FUNCS = {
    "explained_variance_score": explained_variance_score,
    "r2_score": r2_score,
    "mean_squared_error": mean_squared_error,
    "mean_absolute_error": mean_absolute_error,
}


class ModelBuilder:
    def __init__(self, machine: Machine):
        self.machine = machine

    # ------------------------------------------------------------------
    @property
    def cache_key(self) -> str:
        return self.calculate_cache_key(self.machine)

    @staticmethod
    def calculate_cache_key(machine: Machine) -> str:
        payload = json.dumps(
            {
                "name": machine.name,
                "version": __version__,
                "dataset": machine.dataset,
                "model": machine.model,
                "evaluation": machine.evaluation,
            },
            sort_keys=True,
            default=str,
        )
        return hashlib.md5(payload.encode()).hexdigest()

    def check_cache(self, registry_dir: Union[os.PathLike, str]) -> Optional[str]:
        """Previously-built artifact dir for this config hash, if intact."""
        existing = disk_registry.get_value(registry_dir, self.cache_key)
        if existing is None:
            return None
        if not os.path.isdir(existing) or not os.path.exists(
            os.path.join(existing, "model.pkl")
        ):
            logger.warning("Registry entry %s points at a missing dir; rebuilding", existing)
            return None
        return existing

    # ------------------------------------------------------------------
    def build(
        self,
        output_dir: Optional[str] = None,
        model_register_dir: Optional[Union[os.PathLike, str]] = None,
        replace_cache: bool = False,
    ) -> Tuple[Any, Dict[str, Any]]:
        """Build (or reuse) the model; returns (model, metadata).

        When output_dir is given the artifact is dumped there; with
        model_register_dir a cache hit copies/links the cached artifact
        instead of retraining.
        """
        if model_register_dir and not replace_cache:
            cached = self.check_cache(model_register_dir)
            if cached:
                logger.info("Cache hit for %s -> %s", self.machine.name, cached)
                from ..serializer import load, load_metadata

                if output_dir and os.path.abspath(output_dir) != os.path.abspath(cached):
                    os.makedirs(output_dir, exist_ok=True)
                    for item in os.listdir(cached):
                        src = os.path.join(cached, item)
                        dst = os.path.join(output_dir, item)
                        if os.path.isdir(src):
                            shutil.copytree(src, dst, dirs_exist_ok=True)
                        else:
                            shutil.copy2(src, dst)
                return load(cached), load_metadata(cached)

        model, metadata = self._build()

        if output_dir:
            dump(model, output_dir, metadata=metadata)
            if model_register_dir:
                disk_registry.write_key(
                    model_register_dir, self.cache_key, os.path.abspath(output_dir)
                )
        return model, metadata

    # ------------------------------------------------------------------
    def _build(self) -> Tuple[Any, Dict[str, Any]]:
        machine = self.machine
        t_start = time.time()

        dataset = dataset_from_config(machine.dataset)
        t_query = time.time()
        with _phase_range("gordo:data-query"):
            X, y = dataset.get_data()
        data_query_duration = time.time() - t_query

        model = from_definition(machine.model)

        evaluation = machine.evaluation
        cv_mode = evaluation.get("cv_mode", "full_build")
        metadata: Dict[str, Any] = {
            "name": machine.name,
            "version": __version__,
            "dataset": {**dataset.get_metadata(), "data-query-duration-sec": data_query_duration},
            "model": {
                "model-offset": self._model_offset(model),
                "model-config": machine.model,
            },
            "user-defined": machine.metadata,
            "runtime": machine.runtime,
        }

        if cv_mode in ("full_build", "cross_val_only"):
            with _phase_range("gordo:cross-validate"):
                cv_meta = self._cross_validate(model, X, y, evaluation)
            metadata["model"]["cross-validation"] = cv_meta

        if cv_mode != "cross_val_only":
            t_fit = time.time()
            yv = y if y is not None else X
            with _phase_range("gordo:fit"):
                model.fit(_values(X), _values(yv))
            metadata["model"]["model-training-duration-sec"] = time.time() - t_fit
        metadata["model"]["model-builder-version"] = __version__
        if isinstance(model, GordoBase) or hasattr(model, "get_metadata"):
            try:
                metadata["model"]["model-metadata"] = model.get_metadata()
            except Exception:
                pass
        elif hasattr(model, "steps"):
            for _, step in model.steps:
                if hasattr(step, "get_metadata"):
                    metadata["model"]["model-metadata"] = step.get_metadata()
        metadata["model-build-duration-sec"] = time.time() - t_start
        return model, metadata

    # ------------------------------------------------------------------
    def _cross_validate(self, model, X, y, evaluation: Dict[str, Any]) -> Dict[str, Any]:
        t0 = time.time()
        yv = _values(y) if y is not None else _values(X)
        Xv = _values(X)
        n_splits = int(evaluation.get("cv_splits", 3))
        cv = TimeSeriesSplit(n_splits=n_splits)
        metric_names = [
            m
            for m in (evaluation.get("metrics") or ["explained_variance_score"])
            if m in METRIC_FUNCS
        ]

        scoring_scaler = None
        scaler_def = evaluation.get("scoring_scaler")
        if scaler_def:
            scoring_scaler = from_definition(scaler_def)
            scoring_scaler.fit(yv)

        tag_names = [str(c) for c in getattr(X, "columns", range(Xv.shape[1]))]
        per_tag = bool(evaluation.get("per_tag_metrics", True))

        # torch-vectorized fold scoring (K9): one pass per fold covers every
        # (metric, tag) pair — on GPU when present — instead of ~200 sklearn
        # scorer invocations each re-transforming the validation block.
        # Threshold-deriving CV path: only when the TOP-level model is the
        # anomaly detector (wrapping a full inner pipeline) — running the
        # detector's own CV under a partial pipeline would skip outer steps.
        anomaly_target = model if isinstance(model, AnomalyDetectorBase) else None
        cv_kwargs = dict(
            metric_names=metric_names,
            scoring_scaler=scoring_scaler,
            per_tag=per_tag,
            tag_names=tag_names,
        )
        if anomaly_target is not None:
            cv_out = anomaly_target.cross_validate(X=Xv, y=yv, cv=cv, **cv_kwargs)
        else:
            cv_out = run_time_series_cv(model, Xv, yv, cv=cv, **cv_kwargs)
        cv_duration = time.time() - t0

        scores: Dict[str, Any] = {}
        score_keys = [m for m in metric_names]
        if per_tag:
            score_keys += [f"{m}-{t}" for m in metric_names for t in tag_names]
        for mname in score_keys:
            key = f"test_{mname}"
            if key in cv_out:
                vals = [float(v) for v in cv_out[key]]
                scores[mname] = {
                    "folds": vals,
                    "mean": float(np.mean(vals)),
                    "std": float(np.std(vals)),
                    "min": float(np.min(vals)),
                    "max": float(np.max(vals)),
                }
        out = {
            "cv-duration-sec": cv_duration,
            "scores": scores,
            "splits": n_splits,
        }
        if anomaly_target is not None and anomaly_target.aggregate_threshold_ is not None:
            out["thresholds"] = anomaly_target.get_metadata()
        return out

    @staticmethod
    def _find_anomaly_detector(model):
        if isinstance(model, AnomalyDetectorBase):
            return model
        if hasattr(model, "steps"):
            for _, step in model.steps:
                if isinstance(step, AnomalyDetectorBase):
                    return step
        return None

    @staticmethod
    def _model_offset(model) -> int:
        """Rows consumed before the first prediction (lookback models)."""
        lookback = getattr(model, "lookback_window", None)
        if lookback is None and hasattr(model, "steps"):
            for _, step in model.steps:
                lookback = getattr(step, "lookback_window", None)
                if lookback:
                    break
        if not lookback:
            return 0
        lookahead = getattr(model, "lookahead", 0) or 0
        return int(lookback) - 1 + int(lookahead)


def _values(X):
    return X.values if isinstance(X, pd.DataFrame) else np.asarray(X)


def build_model(machine: Machine, **kwargs):
    """Module-level convenience (parity: earlier trees' build_model)."""
    return ModelBuilder(machine).build(**kwargs)
