"""Property-based tests (hypothesis) — ROADMAP verification item 10:
serializer round-trip stability and windower alignment across the
whole parameter space."""
import numpy as np
import pandas as pd
from hypothesis import given, settings
from hypothesis import strategies as st

from gordo_amd import serializer
from gordo_amd.machine.model.models import create_keras_timeseriesgenerator
from gordo_amd.machine.model.utils import trail_min_max


@settings(max_examples=40, deadline=None)
@given(
    rows=st.integers(min_value=1, max_value=60),
    lookback=st.integers(min_value=1, max_value=12),
    lookahead=st.integers(min_value=0, max_value=3),
    batch_size=st.integers(min_value=1, max_value=16),
    n_features=st.integers(min_value=1, max_value=5),
)
def test_windower_alignment_property(rows, lookback, lookahead,
                                     batch_size, n_features):
    """Every produced sample obeys X[j:j+L] / y[j+L-1+lookahead] and
    the sample count is max(0, rows - L + 1 - lookahead)."""
    X = np.arange(rows * n_features, dtype=float).reshape(rows, n_features)
    y = X * 10.0
    gen = create_keras_timeseriesgenerator(
        X, y, batch_size=batch_size, lookback_window=lookback,
        lookahead=lookahead,
    )
    expected = max(0, rows - lookback + 1 - lookahead)
    got = 0
    for b in range(len(gen)):
        bx, by = gen[b]
        for k in range(len(bx)):
            j = got + k
            np.testing.assert_array_equal(bx[k], X[j:j + lookback])
            np.testing.assert_array_equal(
                by[k], y[j + lookback - 1 + lookahead]
            )
        got += len(bx)
    assert got == expected


@settings(max_examples=30, deadline=None)
@given(
    n_components=st.integers(min_value=1, max_value=4),
    whiten=st.booleans(),
    with_scaler=st.booleans(),
)
def test_serializer_roundtrip_property(n_components, whiten, with_scaler):
    """into_definition(from_definition(d)) is a fixed point for
    pipeline definitions over PCA/MinMaxScaler parameter space."""
    steps = []
    if with_scaler:
        steps.append("sklearn.preprocessing.MinMaxScaler")
    steps.append({
        "sklearn.decomposition.PCA": {
            "n_components": n_components, "whiten": whiten,
        }
    })
    definition = {"sklearn.pipeline.Pipeline": {"steps": steps}}
    obj1 = serializer.from_definition(definition)
    d1 = serializer.into_definition(obj1)
    obj2 = serializer.from_definition(d1)
    d2 = serializer.into_definition(obj2)
    assert d1 == d2  # fixed point after one round trip
    pca1 = [s for _, s in obj1.steps][-1]
    assert pca1.n_components == n_components
    assert pca1.whiten == whiten


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(min_value=1, max_value=200),
    w=st.integers(min_value=1, max_value=50),
    data=st.integers(min_value=0, max_value=2 ** 31),
)
def test_trail_min_max_property(n, w, data):
    """trail_min_max == pandas rolling(w).min().max() for any n, w."""
    rng = np.random.default_rng(data)
    a = rng.normal(size=n)
    got = trail_min_max(a, w)
    want = pd.Series(a).rolling(w).min().max()
    if n < w:
        assert np.isnan(got) and np.isnan(want)
    else:
        assert np.isclose(got, want)
