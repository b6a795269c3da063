"""
ModelBuilder — the training engine for one Machine.

Behavioral spec: gordo/builder/build_model.py:48-705 — dataset fetch,
model instantiation from definition, CV scoring with per-tag metrics,
fit, metadata assembly, save (serializer.dump layout) and sha3-512
disk-registry caching. The fleet-scale path (many Machines at once,
grouped across GPUs) lives in ``gordo_amd.parallel``; this class is the
single-machine unit it delegates to, and is also the `gordo build` CLI
entry.
"""
from __future__ import annotations

import datetime
import hashlib
import json
import logging
import os
import random
import time
from pathlib import Path
from typing import Any, Callable, Dict, List, Optional, Tuple, Union

import numpy as np
import pandas as pd
from sklearn import metrics
from sklearn.base import BaseEstimator, TransformerMixin
from sklearn.model_selection import cross_validate
from sklearn.pipeline import Pipeline

import gordo_amd
from .. import serializer
from ..core.base import GordoBaseDataset
from ..machine import Machine
from ..machine.metadata import (
    BuildMetadata,
    CrossValidationMetaData,
    DatasetBuildMetadata,
    ModelBuildMetadata,
)
from ..machine.model.base import GordoBase
from ..machine.model.utils import metric_wrapper
from ..util import disk_registry

logger = logging.getLogger(__name__)


def _has_predict(model) -> bool:
    """hasattr(model, "predict") without sklearn-1.7's unfitted-Pipeline
    FutureWarning (accessing .predict on an unfitted Pipeline warns)."""
    import warnings

    with warnings.catch_warnings():
        warnings.simplefilter("ignore", FutureWarning)
        return hasattr(model, "predict")


class ModelBuilder:
    def __init__(self, machine: Machine):
        self.machine = machine
        self._cached_model_path: Optional[str] = None

    @property
    def gordo_version(self) -> str:
        return gordo_amd.__version__

    @property
    def cached_model_path(self) -> Optional[str]:
        return self._cached_model_path

    def build(
        self,
        output_dir: Optional[Union[os.PathLike, str]] = None,
        model_register_dir: Optional[Union[os.PathLike, str]] = None,
        replace_cache: bool = False,
    ) -> Tuple[Optional[BaseEstimator], Machine]:
        """
        Build the model; if ``model_register_dir`` is given, consult /
        update the build cache keyed by :meth:`calculate_cache_key`.
        """
        if not model_register_dir:
            model, machine = self._build()
            if output_dir:
                self._save_model(model, machine, output_dir)
            return model, machine

        logger.debug(
            "Model register dir: %s; cache key: %s",
            model_register_dir, self.cache_key,
        )
        if replace_cache:
            logger.info("replace_cache=True, deleting any existing cache entry")
            disk_registry.delete_value(model_register_dir, self.cache_key)

        cached_model_path = self.check_cache(model_register_dir, self.cache_key)
        if cached_model_path:
            metadata = serializer.load_metadata(cached_model_path)
            machine = Machine.from_dict(metadata)
            model = serializer.load(cached_model_path)
            self._cached_model_path = cached_model_path
            if output_dir and os.fspath(output_dir) != cached_model_path:
                self._save_model(model, machine, output_dir)
            return model, machine

        model, machine = self._build()
        if output_dir:
            self._save_model(model, machine, output_dir)
            disk_registry.write_key(
                model_register_dir, self.cache_key, os.fspath(output_dir)
            )
            self._cached_model_path = os.fspath(output_dir)
        return model, machine

    # ------------------------------------------------------------------
    def _build(self) -> Tuple[BaseEstimator, Machine]:
        self.set_seed(seed=self.machine.evaluation.get("seed", 0))

        dataset = GordoBaseDataset.from_dict(self.machine.dataset.to_dict())
        start = time.time()
        X, y = dataset.get_data()
        time_elapsed_data = time.time() - start

        model = serializer.from_definition(self.machine.model)
        machine = Machine.from_dict(
            dict(
                name=self.machine.name,
                dataset=self.machine.dataset.to_dict(),
                metadata=self.machine.metadata.to_dict(),
                model=self.machine.model,
                project_name=self.machine.project_name,
                evaluation=self.machine.evaluation,
                runtime=self.machine.runtime,
            )
        )

        cv_duration_sec = None
        split_metadata: Dict[str, Any] = {}
        scores: Dict[str, Any] = {}
        cv_mode = self.machine.evaluation.get("cv_mode", "full_build").lower()
        if cv_mode in ("cross_val_only", "full_build"):
            metrics_list = self.metrics_from_list(
                self.machine.evaluation.get("metrics")
            )
            if _has_predict(model):
                start = time.time()
                scaler = self.machine.evaluation.get("scoring_scaler")
                metrics_dict = self.build_metrics_dict(metrics_list, y, scaler=scaler)
                split_obj = serializer.from_definition(
                    self.machine.evaluation.get(
                        "cv",
                        {"sklearn.model_selection.TimeSeriesSplit": {"n_splits": 3}},
                    )
                )
                split_metadata = self.build_split_dict(X, split_obj)
                cv_kwargs = dict(
                    X=X, y=y, scoring=metrics_dict, return_estimator=True,
                    cv=split_obj,
                )
                if hasattr(model, "cross_validate"):
                    cv = model.cross_validate(**cv_kwargs)
                else:
                    cv = cross_validate(model, **cv_kwargs)

                for metric_name in metrics_dict:
                    arr = cv[f"test_{metric_name}"]
                    val = {
                        "fold-mean": arr.mean(),
                        "fold-std": arr.std(),
                        "fold-max": arr.max(),
                        "fold-min": arr.min(),
                    }
                    val.update(
                        {f"fold-{i + 1}": v for i, v in enumerate(arr.tolist())}
                    )
                    scores[metric_name] = val
                cv_duration_sec = time.time() - start
            else:
                logger.debug("Model has no 'predict'; skipping scoring")

            if cv_mode == "cross_val_only":
                machine.metadata.build_metadata = BuildMetadata(
                    model=ModelBuildMetadata(
                        cross_validation=CrossValidationMetaData(
                            cv_duration_sec=cv_duration_sec,
                            scores=scores,
                            splits=split_metadata,
                        )
                    ),
                    dataset=DatasetBuildMetadata(
                        query_duration_sec=time_elapsed_data,
                        dataset_meta=dataset.get_metadata(),
                    ),
                )
                return model, machine

        start = time.time()
        model.fit(X, y)
        time_elapsed_model = time.time() - start

        machine.metadata.build_metadata = BuildMetadata(
            model=ModelBuildMetadata(
                model_offset=self._determine_offset(model, X),
                model_creation_date=str(
                    datetime.datetime.now(datetime.timezone.utc).astimezone()
                ),
                model_builder_version=self.gordo_version,
                model_training_duration_sec=time_elapsed_model,
                cross_validation=CrossValidationMetaData(
                    cv_duration_sec=cv_duration_sec,
                    scores=scores,
                    splits=split_metadata,
                ),
                model_meta=self._extract_metadata_from_model(model),
            ),
            dataset=DatasetBuildMetadata(
                query_duration_sec=time_elapsed_data,
                dataset_meta=dataset.get_metadata(),
            ),
        )
        return model, machine

    # ------------------------------------------------------------------
    def set_seed(self, seed: int):
        import torch

        logger.info("Setting random seed: %r", seed)
        # Seed the CPU generator ONLY: torch.manual_seed also resets
        # every CUDA generator's state, which orphans live captured
        # hipGraphs (their destructors then throw c10::Error
        # "The graph should be registered to the state" inside GC and
        # terminate the process — HIPGeneratorImpl.cpp:158). The engine
        # draws all randomness from explicit CPU generators.
        torch.default_generator.manual_seed(seed)
        np.random.seed(seed)
        random.seed(seed)

    @staticmethod
    def build_split_dict(X: pd.DataFrame, split_obj) -> dict:
        split_metadata: Dict[str, Any] = {}
        for i, (train_ind, test_ind) in enumerate(split_obj.split(X)):
            split_metadata.update(
                {
                    f"fold-{i + 1}-train-start": X.index[train_ind[0]],
                    f"fold-{i + 1}-train-end": X.index[train_ind[-1]],
                    f"fold-{i + 1}-test-start": X.index[test_ind[0]],
                    f"fold-{i + 1}-test-end": X.index[test_ind[-1]],
                    f"fold-{i + 1}-n-train": len(train_ind),
                    f"fold-{i + 1}-n-test": len(test_ind),
                }
            )
        return split_metadata

    @staticmethod
    def build_metrics_dict(
        metrics_list: list,
        y: pd.DataFrame,
        scaler: Optional[Union[TransformerMixin, str, dict]] = None,
    ) -> dict:
        """Per-tag scorers named '{metric}-{tag}' plus the aggregate
        '{metric}' (reference build_model.py:378-446)."""
        if scaler:
            if isinstance(scaler, (str, dict)):
                if isinstance(scaler, str):
                    scaler = {scaler: {}}
                scaler = serializer.from_definition(scaler)
            scaler.fit(y)

        def _score_factory(metric_func, col_index):
            def _score_per_tag(y_true, y_pred):
                y_true = getattr(y_true, "values", y_true)
                y_pred = getattr(y_pred, "values", y_pred)
                return metric_func(y_true[:, col_index], y_pred[:, col_index])

            return _score_per_tag

        metrics_dict = {}
        for metric in metrics_list:
            metric_str = metric.__name__.replace("_", "-")
            for index, col in enumerate(y.columns):
                metrics_dict[
                    metric_str + f'-{str(col).replace(" ", "-")}'
                ] = metrics.make_scorer(
                    metric_wrapper(
                        _score_factory(metric, index), scaler=scaler
                    )
                )
            metrics_dict[metric_str] = metrics.make_scorer(
                metric_wrapper(metric, scaler=scaler)
            )
        return metrics_dict

    @staticmethod
    def _determine_offset(model: BaseEstimator, X) -> int:
        X = getattr(X, "values", X)
        out = model.predict(X) if _has_predict(model) else model.transform(X)
        return len(X) - len(out)

    @staticmethod
    def _save_model(model: BaseEstimator, machine: Machine, output_dir):
        os.makedirs(output_dir, exist_ok=True)
        machine.metadata.user_defined["model-builder"] = {
            "version": gordo_amd.__version__,
        }
        serializer.dump(
            model, os.fspath(output_dir),
            metadata=json.loads(machine.to_json()),
            info={},
        )

    @staticmethod
    def _extract_metadata_from_model(
        model: BaseEstimator, metadata: Optional[dict] = None
    ) -> dict:
        """Recurse the estimator graph bottom-up collecting
        get_metadata() dicts (reference build_model.py:516-573)."""
        metadata = dict(metadata) if metadata is not None else {}
        # a Pipeline's metadata is its final step's
        if isinstance(model, Pipeline):
            metadata.update(
                ModelBuilder._extract_metadata_from_model(model.steps[-1][1])
            )
            return metadata
        if isinstance(model, GordoBase):
            metadata.update(model.get_metadata())
        # keep recursing into attributes: a GordoBase (e.g. the anomaly
        # detector) may hold another estimator (its base_estimator) whose
        # metadata must also surface (reference build_model.py:552-573)
        for key, val in vars(model).items():
            if key == "regressor":
                continue
            if isinstance(val, Pipeline):
                metadata.update(
                    ModelBuilder._extract_metadata_from_model(val.steps[-1][1])
                )
            elif isinstance(val, (GordoBase, BaseEstimator)):
                metadata.update(ModelBuilder._extract_metadata_from_model(val))
        return metadata

    # ------------------------------------------------------------------
    @property
    def cache_key(self) -> str:
        return self.calculate_cache_key(self.machine)

    def calculate_cache_key(self, machine: Machine) -> str:
        """
        sha3-512 over the sorted-JSON of {name, model config, dataset
        config, evaluation config, framework major/minor (+full version
        when unstable)} (reference build_model.py:575-631).

        >>> from gordo_amd.machine import Machine
        >>> machine = Machine.from_config(dict(
        ...     name="special-model-name",
        ...     model={"sklearn.decomposition.PCA": {"svd_solver": "auto"}},
        ...     dataset={
        ...         "type": "RandomDataset",
        ...         "train_start_date": "2017-12-25 06:00:00Z",
        ...         "train_end_date": "2017-12-30 06:00:00Z",
        ...         "tag_list": ["Tag 1", "Tag 2"],
        ...     },
        ... ), project_name="test-proj")
        >>> len(ModelBuilder(machine).cache_key)
        128
        """
        from gordo_amd import parse_version

        major, minor, unstable = parse_version(self.gordo_version)
        json_rep = json.dumps(
            {
                "name": machine.name,
                "model_config": machine.model,
                "data_config": machine.dataset.to_dict(),
                "evaluation_config": machine.evaluation,
                "gordo-major-version": major,
                "gordo-minor-version": minor,
                "gordo_version": self.gordo_version if unstable else "",
            },
            sort_keys=True,
            default=str,
        )
        return hashlib.sha3_512(json_rep.encode("ascii")).hexdigest()

    @staticmethod
    def check_cache(model_register_dir, cache_key: str) -> Optional[str]:
        existing = disk_registry.get_value(model_register_dir, cache_key)
        if existing and Path(existing).exists():
            return existing
        if existing:
            logger.warning(
                "Registered model path %s no longer exists", existing
            )
        return None

    @staticmethod
    def metrics_from_list(
        metric_list: Optional[List[str]] = None,
    ) -> List[Callable]:
        """
        Load metric callables from dotted paths or bare sklearn.metrics
        names.

        >>> [m.__name__ for m in ModelBuilder.metrics_from_list()]
        ['explained_variance_score', 'r2_score', 'mean_squared_error', 'mean_absolute_error']
        """
        if metric_list is None:
            metric_list = [
                "sklearn.metrics.explained_variance_score",
                "sklearn.metrics.r2_score",
                "sklearn.metrics.mean_squared_error",
                "sklearn.metrics.mean_absolute_error",
            ]
        from ..core.import_utils import import_location

        out = []
        for m in metric_list:
            if "." not in m:
                m = f"sklearn.metrics.{m}"
            out.append(import_location(m))
        return out
