"""
Response-frame assembly and metric helpers
(behavioral spec: gordo/machine/model/utils.py:18-165).
"""
from __future__ import annotations

import functools
import logging
from datetime import timedelta
from typing import List, Optional, Union

import numpy as np
import pandas as pd

from ...core.sensor_tag import SensorTag

logger = logging.getLogger(__name__)


def metric_wrapper(metric, scaler=None):
    """Wrap a metric so y_true is trimmed to the model-output length
    (lookback offset) and both sides optionally pass through a fitted
    scaler first."""

    @functools.wraps(metric)
    def _wrapper(y_true, y_pred, *args, **kwargs):
        if scaler:
            y_true = scaler.transform(y_true)
            y_pred = scaler.transform(y_pred)
        return metric(y_true[-len(y_pred):], y_pred, *args, **kwargs)

    return _wrapper


def _tag_name(tag) -> str:
    return tag.name if isinstance(tag, SensorTag) else str(tag)


_START_END_COLS = pd.MultiIndex.from_product((("start", "end"), ("",)))


@functools.lru_cache(maxsize=1024)
def _mi_tuples(name: str, sub_names: tuple) -> pd.MultiIndex:
    """Static per (family, tag list) — cached; ~ms-scale pandas
    factorize per construction on the serving path."""
    return pd.MultiIndex.from_tuples((name, s) for s in sub_names)


def make_base_dataframe(
    tags: Union[List[SensorTag], List[str]],
    model_input: np.ndarray,
    model_output: np.ndarray,
    target_tag_list: Optional[Union[List[SensorTag], List[str]]] = None,
    index: Optional[Union[np.ndarray, pd.Index]] = None,
    frequency: Optional[timedelta] = None,
) -> pd.DataFrame:
    """
    Build the MultiIndex response frame with top-level column groups
    ``start``/``end``/``model-input``/``model-output``, aligning the
    input and index to the (possibly offset) model-output length and
    ISO-formatting timestamps.
    """
    target_tag_list = target_tag_list if target_tag_list is not None else tags

    model_input = getattr(model_input, "values", model_input)
    model_output = getattr(model_output, "values", model_output)
    n_out = len(model_output)
    model_input = model_input[-n_out:, :]

    if index is not None:
        normalized_index = pd.Index(index[-n_out:])
    else:
        normalized_index = pd.RangeIndex(n_out)

    if isinstance(normalized_index, pd.DatetimeIndex):
        starts = [ts.isoformat() for ts in normalized_index]
        if frequency is not None:
            ends = [(ts + frequency).isoformat() for ts in normalized_index]
        else:
            ends = [None] * n_out
    else:
        starts = [None] * n_out
        ends = [None] * n_out

    blocks = [
        pd.DataFrame(
            {("start", ""): starts, ("end", ""): ends},
            columns=_START_END_COLS,
            index=normalized_index,
        )
    ]

    for name, values, _tags in (
        ("model-input", model_input, tags),
        ("model-output", model_output, target_tag_list),
    ):
        if values is None:
            continue
        if values.shape[1] == len(_tags):
            sub_names = [_tag_name(t) for t in _tags]
        else:
            sub_names = [str(i) for i in range(values.shape[1])]
        columns = _mi_tuples(name, tuple(sub_names))
        blocks.append(
            pd.DataFrame(values, columns=columns, index=normalized_index)
        )
    # one concat, not sequential joins (each join re-indexes the frame;
    # this is the serving hot path — reference server flow SURVEY §3.2)
    return pd.concat(blocks, axis=1, copy=False)


def trail_min_max(x, w: int):
    """``pd.Series/DataFrame(x).rolling(w).min().max()`` in O(n) C time
    (scipy minimum_filter1d; pandas rolling is O(n*w) and dominated the
    fleet build's CV threshold phase). origin=(w-1)//2 turns the
    centered filter into the trailing window; the first w-1 positions
    (NaN under pandas) are sliced off before the max — exactly pandas'
    NaN-skipping max. Exact-equivalence tested in
    tests/test_packed.py::test_trail_min_max_matches_pandas.

    >>> import numpy as np, pandas as pd
    >>> a = np.array([5.0, 1.0, 4.0, 2.0, 8.0, 0.5])
    >>> float(trail_min_max(a, 3))
    2.0
    >>> float(pd.Series(a).rolling(3).min().max())
    2.0
    """
    import numpy as np
    from scipy.ndimage import minimum_filter1d

    x = np.asarray(x)
    if x.shape[0] < w:
        return np.nan if x.ndim == 1 else np.full(x.shape[1], np.nan)
    mf = minimum_filter1d(x, size=w, axis=0, mode="nearest", origin=(w - 1) // 2)
    return mf[w - 1:].max(axis=0)
