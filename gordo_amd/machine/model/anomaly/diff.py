"""
Diff-based anomaly detectors.

Behavioral spec: gordo/machine/model/anomaly/diff.py —
``DiffBasedAnomalyDetector`` (:21) computes residual-based anomaly
frames with thresholds from TimeSeriesSplit CV rolling statistics;
``DiffBasedKFCVAnomalyDetector`` (:461) uses KFold CV and a percentile
of the smoothed validation metric. Column families and threshold math
match the reference exactly (tests assert them); the serving-path
residual computation runs on device through the fused scoring op when
the model output is resident (kernel K9 of SURVEY.md §2.3).
"""
from __future__ import annotations

import functools
import logging
from datetime import timedelta
from typing import Optional, Union

import numpy as np
import pandas as pd
from sklearn.base import BaseEstimator, TransformerMixin
from sklearn.exceptions import NotFittedError
from sklearn.metrics import explained_variance_score
from sklearn.model_selection import KFold, TimeSeriesSplit
from sklearn.model_selection import cross_validate as c_val
from sklearn.preprocessing import MinMaxScaler
from sklearn.utils import shuffle as sk_shuffle

from .base import AnomalyDetectorBase
from ..base import GordoBase
from .. import utils as model_utils

logger = logging.getLogger(__name__)


def _default_base_estimator():
    from ..models import KerasAutoEncoder

    return KerasAutoEncoder(kind="feedforward_hourglass")


class DiffBasedAnomalyDetector(AnomalyDetectorBase):
    def __init__(
        self,
        base_estimator: Optional[BaseEstimator] = None,
        scaler: Optional[TransformerMixin] = None,
        require_thresholds: bool = True,
        shuffle: bool = False,
        window: Optional[int] = None,
        smoothing_method: Optional[str] = None,
    ):
        self.base_estimator = (
            base_estimator if base_estimator is not None else _default_base_estimator()
        )
        self.scaler = scaler if scaler is not None else MinMaxScaler()
        self.require_thresholds = require_thresholds
        self.shuffle = shuffle
        self.window = window
        self.smoothing_method = smoothing_method
        if self.window is not None and self.smoothing_method is None:
            self.smoothing_method = "smm"

    def __getattr__(self, item):
        # transparent passthrough into base_estimator (reference diff.py:78)
        if item.startswith("__") and item.endswith("__"):
            raise AttributeError(item)
        base = self.__dict__.get("base_estimator")
        if base is None:
            raise AttributeError(item)
        return getattr(base, item)

    def get_params(self, deep=True):
        params = {
            "base_estimator": self.base_estimator,
            "scaler": self.scaler,
            "require_thresholds": self.require_thresholds,
            "shuffle": self.shuffle,
        }
        if self.window is not None:
            params["window"] = self.window
            params["smoothing_method"] = self.smoothing_method
        return params

    def get_metadata(self):
        metadata = {}
        if hasattr(self, "feature_thresholds_"):
            metadata["feature-thresholds"] = self.feature_thresholds_.tolist()
        if hasattr(self, "aggregate_threshold_"):
            metadata["aggregate-threshold"] = self.aggregate_threshold_
        if hasattr(self, "feature_thresholds_per_fold_"):
            metadata["feature-thresholds-per-fold"] = (
                self.feature_thresholds_per_fold_.to_dict()
            )
        if hasattr(self, "aggregate_thresholds_per_fold_"):
            metadata["aggregate-thresholds-per-fold"] = (
                self.aggregate_thresholds_per_fold_
            )
        metadata["window"] = self.window
        metadata["smoothing-method"] = self.smoothing_method
        if (
            hasattr(self, "smooth_feature_thresholds_")
            and self.smooth_feature_thresholds_ is not None
        ):
            metadata["smooth-feature-thresholds"] = (
                self.smooth_feature_thresholds_.tolist()
            )
        if (
            hasattr(self, "smooth_aggregate_threshold_")
            and self.smooth_aggregate_threshold_ is not None
        ):
            metadata["smooth-aggregate-threshold"] = self.smooth_aggregate_threshold_
        if (
            getattr(self, "smooth_feature_thresholds_per_fold_", None) is not None
            and len(self.smooth_feature_thresholds_per_fold_) > 0
        ):
            metadata["smooth-feature-thresholds-per-fold"] = (
                self.smooth_feature_thresholds_per_fold_.to_dict()
            )
        if getattr(self, "smooth_aggregate_thresholds_per_fold_", None):
            metadata["smooth-aggregate-thresholds-per-fold"] = (
                self.smooth_aggregate_thresholds_per_fold_
            )
        if isinstance(self.base_estimator, GordoBase):
            metadata.update(self.base_estimator.get_metadata())
        else:
            metadata.update(
                {
                    "scaler": str(self.scaler),
                    "base_estimator": str(self.base_estimator),
                    "shuffle": self.shuffle,
                }
            )
        return metadata

    def score(self, X, y=None, sample_weight=None):
        y = X if y is None else y
        out = self.base_estimator.predict(np.asarray(getattr(X, "values", X)))
        y_arr = np.asarray(getattr(y, "values", y))
        return explained_variance_score(y_arr[-len(out):], out)

    def fit(self, X, y=None):
        if y is None:
            y = X
        if self.shuffle:
            X_shuff, y_shuff = sk_shuffle(X, y, random_state=0)
            self.base_estimator.fit(X_shuff, y_shuff)
        else:
            self.base_estimator.fit(X, y)
        self.scaler.fit(y)  # used only for error calculations in .anomaly()
        return self

    def cross_validate(
        self,
        *,
        X: Union[pd.DataFrame, np.ndarray],
        y: Union[pd.DataFrame, np.ndarray],
        cv=None,
        **kwargs,
    ):
        if cv is None:
            cv = TimeSeriesSplit(n_splits=3)
        kwargs.update(dict(return_estimator=True, cv=cv))
        cv_output = c_val(self, X=X, y=y, **kwargs)

        self.feature_thresholds_per_fold_ = pd.DataFrame()
        self.aggregate_thresholds_per_fold_ = {}
        self.smooth_feature_thresholds_per_fold_ = pd.DataFrame()
        self.smooth_aggregate_thresholds_per_fold_ = {}
        aggregate_threshold_fold = None
        tag_thresholds_fold = None
        smooth_aggregate_threshold_fold = None
        smooth_tag_thresholds_fold = None

        for i, ((_, test_idxs), split_model) in enumerate(
            zip(kwargs["cv"].split(X, y), cv_output["estimator"])
        ):
            y_pred = split_model.predict(
                X.iloc[test_idxs] if isinstance(X, pd.DataFrame) else X[test_idxs]
            )
            test_idxs = test_idxs[-len(y_pred):]
            y_true = y.iloc[test_idxs] if isinstance(y, pd.DataFrame) else y[test_idxs]

            scaled_mse = self._scaled_mse_per_timestep(split_model, y_true, y_pred)
            mae = self._absolute_error(y_true, y_pred)

            aggregate_threshold_fold = model_utils.trail_min_max(
                scaled_mse.to_numpy(), 6
            )
            self.aggregate_thresholds_per_fold_[f"fold-{i}"] = aggregate_threshold_fold

            tag_thresholds_fold = pd.Series(
                model_utils.trail_min_max(mae.to_numpy(), 6),
                index=mae.columns,
                name=f"fold-{i}",
            )
            self.feature_thresholds_per_fold_ = pd.concat(
                [self.feature_thresholds_per_fold_, tag_thresholds_fold.to_frame().T]
            )

            if self.window is not None:
                smooth_aggregate_threshold_fold = (
                    model_utils.trail_min_max(scaled_mse.to_numpy(), self.window)
                )
                self.smooth_aggregate_thresholds_per_fold_[f"fold-{i}"] = (
                    smooth_aggregate_threshold_fold
                )
                smooth_tag_thresholds_fold = pd.Series(
                    model_utils.trail_min_max(mae.to_numpy(), self.window),
                    index=mae.columns,
                )
                smooth_tag_thresholds_fold.name = f"fold-{i}"
                self.smooth_feature_thresholds_per_fold_ = pd.concat(
                    [
                        self.smooth_feature_thresholds_per_fold_,
                        smooth_tag_thresholds_fold.to_frame().T,
                    ]
                )

        # thresholds come from the last fold (reference diff.py:256-264)
        self.feature_thresholds_ = tag_thresholds_fold
        self.aggregate_threshold_ = aggregate_threshold_fold
        self.smooth_aggregate_threshold_ = smooth_aggregate_threshold_fold
        self.smooth_feature_thresholds_ = smooth_tag_thresholds_fold
        return cv_output

    @staticmethod
    def _scaled_mse_per_timestep(model, y_true, y_pred) -> pd.Series:
        try:
            scaled_y_true = model.scaler.transform(y_true)
        except (NotFittedError, ValueError):
            scaled_y_true = model.scaler.fit_transform(y_true)
        scaled_y_pred = model.scaler.transform(y_pred)
        mse_per_time_step = ((scaled_y_pred - scaled_y_true) ** 2).mean(axis=1)
        return pd.Series(np.asarray(mse_per_time_step))

    @staticmethod
    def _absolute_error(y_true, y_pred) -> pd.DataFrame:
        return pd.DataFrame(np.abs(np.asarray(y_true) - np.asarray(y_pred)))

    def _smoothing(self, metric):
        if self.smoothing_method == "smm":
            return metric.rolling(self.window).median()
        if self.smoothing_method == "sma":
            return metric.rolling(self.window).mean()
        if self.smoothing_method == "ewma":
            return metric.ewm(span=self.window).mean()
        raise ValueError(f"Unknown smoothing_method {self.smoothing_method!r}")

    # rows below which the H2D round-trip outweighs the fused kernel
    _DEVICE_SCORE_MIN_ROWS = 512

    def set_serving_device(self, device: str):
        """Pin the fused scoring kernel (and, via the server's object
        walk, the nested estimator) to ``device`` for multi-GPU
        serving."""
        self._serve_device = device

    def _fused_device_scores(self, data: pd.DataFrame, y: pd.DataFrame):
        """Serving hot path: one fused HIP kernel (ops.anomaly_score,
        kernel K9) computes every residual column family when a GPU is
        present and the scaler is a fitted MinMaxScaler-style affine
        transform. Returns (tag_scaled, total_scaled, tag_unscaled,
        total_unscaled) as numpy, or None to use the pandas path."""
        try:
            import torch

            from .... import ops

            if (
                len(data) < self._DEVICE_SCORE_MIN_ROWS
                or not torch.cuda.is_available()
                or not ops.hip_available()
                or not hasattr(self.scaler, "scale_")
                or not hasattr(self.scaler, "min_")
            ):
                return None
            device = getattr(self, "_serve_device", None) or "cuda"
            out = torch.as_tensor(
                np.ascontiguousarray(
                    data["model-output"].to_numpy(dtype=np.float32)
                ),
                device=device,
            )
            yt = torch.as_tensor(
                np.ascontiguousarray(
                    y.to_numpy(dtype=np.float32)[-len(data):, :]
                ),
                device=device,
            )
            scale = torch.as_tensor(
                np.asarray(self.scaler.scale_, dtype=np.float32), device=device
            )
            minv = torch.as_tensor(
                np.asarray(self.scaler.min_, dtype=np.float32), device=device
            )
            ts, tots, tu, totu, _, _ = ops.anomaly_score(
                out, yt, scale, minv, None, 1.0
            )
            return (
                ts.cpu().numpy(),
                tots.cpu().numpy(),
                tu.cpu().numpy(),
                totu.cpu().numpy(),
            )
        except Exception:  # any wobble → exact pandas path
            logger.debug("device anomaly scoring unavailable", exc_info=True)
            return None

    def anomaly(
        self,
        X: pd.DataFrame,
        y: pd.DataFrame,
        frequency: Optional[timedelta] = None,
    ) -> pd.DataFrame:
        if not hasattr(X, "values"):
            raise ValueError("Unable to find X.values property")

        model_output = (
            self.predict(X) if hasattr(self, "predict") else self.transform(X)
        )

        data = model_utils.make_base_dataframe(
            tags=X.columns,
            model_input=X.values,
            model_output=model_output,
            target_tag_list=y.columns,
            index=getattr(X, "index", None),
            frequency=frequency,
        )

        scored = self._fused_device_scores(data, y)
        if scored is not None:
            ts, tots, tu, totu = scored
        else:
            model_out_scaled = self.scaler.transform(data["model-output"])
            scaled_y = self.scaler.transform(y)
            ts = np.abs(model_out_scaled - scaled_y[-len(data):, :])
            tots = np.square(ts).mean(axis=1)
            tu = np.abs(
                data["model-output"].to_numpy() - y.to_numpy()[-len(data):, :]
            )
            totu = np.square(tu).mean(axis=1)

        # assemble every column family in ONE concat: sequential
        # MultiIndex .join()s re-index the growing frame per family and
        # were the dominant pandas cost of the serving request
        # (column names/order identical to the reference's join chain,
        # reference diff.py:356-444)
        y_cols = y.columns.tolist()
        index = data.index

        def fam(name, values, cols=y_cols):
            return pd.DataFrame(
                values,
                index=index,
                columns=_mi_product(name, tuple(cols)),
            )

        def fam1(name, values):
            # scalar families carry an empty sub-level, exactly like
            # `frame[name] = values` on a 2-level frame
            return pd.DataFrame(
                np.asarray(values).reshape(-1, 1),
                index=index,
                columns=_mi_product(name, ("",)),
            )

        tag_anomaly_scaled = fam("tag-anomaly-scaled", ts)
        unscaled_abs_diff = fam("tag-anomaly-unscaled", tu)
        blocks = [
            data,
            tag_anomaly_scaled,
            fam1("total-anomaly-scaled", tots),
            unscaled_abs_diff,
            fam1("total-anomaly-unscaled", totu),
        ]

        if self.window is not None and self.smoothing_method is not None:
            smooth_tag_anomaly_scaled = self._smoothing(tag_anomaly_scaled)
            smooth_tag_anomaly_scaled.columns = (
                smooth_tag_anomaly_scaled.columns.set_levels(
                    ["smooth-tag-anomaly-scaled"], level=0
                )
            )
            blocks.append(smooth_tag_anomaly_scaled)
            blocks.append(
                fam1(
                    "smooth-total-anomaly-scaled",
                    self._smoothing(pd.Series(tots, index=index)),
                )
            )
            smooth_tag_anomaly_unscaled = self._smoothing(unscaled_abs_diff)
            smooth_tag_anomaly_unscaled.columns = (
                smooth_tag_anomaly_unscaled.columns.set_levels(
                    ["smooth-tag-anomaly-unscaled"], level=0
                )
            )
            blocks.append(smooth_tag_anomaly_unscaled)
            blocks.append(
                fam1(
                    "smooth-total-anomaly-unscaled",
                    self._smoothing(pd.Series(totu, index=index)),
                )
            )

        if hasattr(self, "feature_thresholds_"):
            confidence = tu / self.feature_thresholds_.values
            blocks.append(
                fam(
                    "anomaly-confidence",
                    confidence,
                    data["model-output"].columns,
                )
            )

        if hasattr(self, "aggregate_threshold_"):
            blocks.append(
                fam1(
                    "total-anomaly-confidence",
                    tots / self.aggregate_threshold_,
                )
            )
        data = pd.concat(blocks, axis=1, copy=False)

        if self.require_thresholds and not any(
            hasattr(self, attr)
            for attr in ("feature_thresholds_", "aggregate_threshold_")
        ):
            raise AttributeError(
                f"`require_thresholds={self.require_thresholds}` however "
                "`.cross_validate` needs to be called in order to calculate "
                "these thresholds before calling `.anomaly`"
            )
        return data


@functools.lru_cache(maxsize=1024)
def _mi_product(name: str, cols: tuple) -> pd.MultiIndex:
    """Column MultiIndexes are static per (family, tag list) — building
    them per request cost ~8 ms/call of the serving path (pandas
    factorize); MultiIndex objects are immutable, safe to share."""
    return pd.MultiIndex.from_product(((name,), list(cols)))


class DiffBasedKFCVAnomalyDetector(DiffBasedAnomalyDetector):
    def __init__(
        self,
        base_estimator: Optional[BaseEstimator] = None,
        scaler: Optional[TransformerMixin] = None,
        require_thresholds: bool = True,
        shuffle: bool = True,
        window: int = 144,
        smoothing_method: str = "smm",
        threshold_percentile: float = 0.99,
    ):
        self.base_estimator = (
            base_estimator if base_estimator is not None else _default_base_estimator()
        )
        self.scaler = scaler if scaler is not None else MinMaxScaler()
        self.require_thresholds = require_thresholds
        self.window = window
        self.shuffle = shuffle
        self.smoothing_method = smoothing_method
        self.threshold_percentile = threshold_percentile

    def get_params(self, deep=True):
        return {
            "base_estimator": self.base_estimator,
            "scaler": self.scaler,
            "window": self.window,
            "smoothing_method": self.smoothing_method,
            "shuffle": self.shuffle,
            "threshold_percentile": self.threshold_percentile,
        }

    def get_metadata(self):
        metadata = {}
        if hasattr(self, "feature_thresholds_"):
            metadata["feature-thresholds"] = self.feature_thresholds_.tolist()
        if hasattr(self, "aggregate_threshold_"):
            metadata["aggregate-threshold"] = self.aggregate_threshold_
        if isinstance(self.base_estimator, GordoBase):
            metadata.update(self.base_estimator.get_metadata())
        else:
            metadata.update(
                {
                    "scaler": str(self.scaler),
                    "base_estimator": str(self.base_estimator),
                    "shuffle": self.shuffle,
                    "window": self.window,
                    "smoothing-method": self.smoothing_method,
                    "threshold-percentile": self.threshold_percentile,
                }
            )
        return metadata

    def cross_validate(
        self,
        *,
        X: Union[pd.DataFrame, np.ndarray],
        y: Union[pd.DataFrame, np.ndarray],
        cv=None,
        **kwargs,
    ):
        if cv is None:
            cv = KFold(n_splits=5, shuffle=True, random_state=0)
        kwargs.update(dict(return_estimator=True, cv=cv))
        cv_output = c_val(self, X=X, y=y, **kwargs)

        y = pd.DataFrame(y)
        y_pred = pd.DataFrame(
            np.zeros_like(y), index=y.index, columns=y.columns
        )
        y_val_mse = pd.Series(np.nan, index=y.index, dtype=float)

        for i, ((_, test_idxs), split_model) in enumerate(
            zip(kwargs["cv"].split(X, y), cv_output["estimator"])
        ):
            y_pred.iloc[test_idxs] = split_model.predict(
                X.iloc[test_idxs].to_numpy()
                if isinstance(X, pd.DataFrame)
                else X[test_idxs]
            )
            y_val_mse.iloc[test_idxs] = self._scaled_mse_per_timestep(
                split_model, y.iloc[test_idxs], y_pred.iloc[test_idxs]
            ).to_numpy()

        self.aggregate_threshold_ = self._calculate_threshold(y_val_mse)
        self.feature_thresholds_ = self._calculate_feature_thresholds(y, y_pred)
        return cv_output

    def _calculate_feature_thresholds(self, y_true, y_pred):
        return self._calculate_threshold(self._absolute_error(y_true, y_pred))

    def _calculate_threshold(self, validation_metric):
        dev = self._device_threshold(validation_metric)
        if dev is not None:
            return dev
        return self._smoothing(validation_metric).quantile(
            self.threshold_percentile
        )

    def _device_threshold(self, validation_metric):
        """K11+K12 device path: rolling(window).median() then
        NaN-dropping quantile, both as HIP kernels when a GPU is
        present and the smoothing is the default smm. Matches the
        pandas pipeline at fp32 (min_periods=window NaN semantics in
        the windowed kernel; linear interpolation in both). Returns a
        Series for DataFrame input, a float for Series input, or None
        to take the pandas path."""
        try:
            import torch

            from .... import ops

            if (
                self.smoothing_method != "smm"
                or not torch.cuda.is_available()
                or not ops.hip_available()
                or len(validation_metric) < max(self.window, 64)
                or len(validation_metric) > 16384
            ):
                return None
            arr = np.asarray(validation_metric, dtype=np.float32)
            cols = arr.reshape(len(arr), -1).T  # [R, N]
            dev = torch.as_tensor(np.ascontiguousarray(cols),
                                  device="cuda")
            sm = ops.windowed_quantile(dev, int(self.window), 0.5)
            q = ops.row_quantile(sm, float(self.threshold_percentile))
            out = q.cpu().numpy().astype(np.float64)
            if isinstance(validation_metric, pd.DataFrame):
                return pd.Series(out, index=validation_metric.columns,
                                 name=self.threshold_percentile)
            return float(out[0])
        except Exception:
            logger.debug("device threshold unavailable", exc_info=True)
            return None
