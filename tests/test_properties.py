"""Property-based tests (hypothesis) for the invariants the framework
leans on hardest: serializer round-trips, windower alignment, and
patch_dict merge semantics."""
import string

import numpy as np
from hypothesis import given, settings, strategies as st

from gordo_amd.machine.model.models import create_keras_timeseriesgenerator
from gordo_amd.serializer import from_definition, into_definition
from gordo_amd.workflow.workflow_generator.helpers import patch_dict

keys = st.text(string.ascii_lowercase, min_size=1, max_size=5)
scalars = st.one_of(st.integers(-100, 100), st.booleans(),
                    st.text(string.ascii_letters, max_size=8))
nested = st.recursive(
    scalars, lambda inner: st.dictionaries(keys, inner, max_size=4),
    max_leaves=12,
)


@given(st.dictionaries(keys, nested, max_size=5),
       st.dictionaries(keys, nested, max_size=5))
@settings(max_examples=60, deadline=None)
def test_patch_dict_properties(original, patch):
    merged = patch_dict(original, patch)
    # never removes keys
    assert set(merged) >= set(original)
    assert set(merged) >= set(patch)
    # non-dict patch values always win
    for k, v in patch.items():
        if not isinstance(v, dict) or not isinstance(original.get(k), dict):
            assert merged[k] == v
    # inputs not mutated
    assert patch_dict(original, {}) == original


@given(
    rows=st.integers(5, 200),
    lookback=st.integers(1, 20),
    lookahead=st.integers(0, 3),
    batch=st.integers(1, 32),
)
@settings(max_examples=60, deadline=None)
def test_windower_alignment_property(rows, lookback, lookahead, batch):
    if lookback + lookahead >= rows:
        return
    X = np.arange(rows, dtype="float64").reshape(rows, 1)
    gen = create_keras_timeseriesgenerator(X, X, batch, lookback, lookahead)
    total = sum(len(gen[i][0]) for i in range(len(gen)))
    assert total == rows - lookback + 1 - lookahead
    # every sample: window is contiguous and the target trails the
    # window end by exactly `lookahead`
    for i in range(len(gen)):
        bx, by = gen[i]
        for j in range(len(bx)):
            start = int(bx[j][0][0])
            assert list(bx[j][:, 0]) == list(range(start, start + lookback))
            assert by[j][0] == start + lookback - 1 + lookahead


@given(
    n_components=st.integers(1, 5),
    with_scaler=st.booleans(),
)
@settings(max_examples=30, deadline=None)
def test_serializer_roundtrip_property(n_components, with_scaler):
    steps = []
    if with_scaler:
        steps.append("sklearn.preprocessing.MinMaxScaler")
    steps.append(
        {"sklearn.decomposition.PCA": {"n_components": n_components}}
    )
    definition = {"sklearn.pipeline.Pipeline": {"steps": steps}}
    model = from_definition(definition)
    back = into_definition(model)
    model2 = from_definition(back)
    assert into_definition(model2) == back
    assert model2.steps[-1][1].n_components == n_components
