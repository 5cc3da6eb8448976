"""Property-based robustness tests for the GBT engine (hypothesis)."""

import numpy as np
from hypothesis import given, settings, strategies as st
from hypothesis.extra.numpy import arrays

from sparkdl.xgboost import gbt


@st.composite
def matrices(draw, max_n=80, max_f=6):
    n = draw(st.integers(8, max_n))
    f = draw(st.integers(1, max_f))
    X = draw(arrays(np.float64, (n, f),
                    elements=st.floats(-100, 100, allow_nan=False)))
    return X


@settings(max_examples=25, deadline=None)
@given(matrices())
def test_binner_roundtrip_ordering(X):
    """Binned codes are monotone in the raw value per feature."""
    binner = gbt.Binner().fit(X)
    B = binner.transform(X)
    for f in range(X.shape[1]):
        order = np.argsort(X[:, f], kind="stable")
        bins_sorted = B[order, f].astype(int)
        assert (np.diff(bins_sorted) >= 0).all()
        assert (B[:, f] != gbt.MISSING_BIN).all()


@settings(max_examples=25, deadline=None)
@given(matrices(), st.floats(0, 1))
def test_missing_always_missing_bin(X, frac):
    rng = np.random.RandomState(0)
    mask = rng.rand(*X.shape) < frac * 0.5
    X = X.copy()
    X[mask] = np.nan
    binner = gbt.Binner().fit(X)
    B = binner.transform(X)
    assert (B[mask] == gbt.MISSING_BIN).all()
    assert (B[~mask] != gbt.MISSING_BIN).all()


@settings(max_examples=10, deadline=None)
@given(matrices(max_n=60, max_f=4),
       st.integers(1, 4), st.integers(1, 3))
def test_train_predict_finite(X, rounds, depth):
    rng = np.random.RandomState(1)
    y = rng.randn(X.shape[0])
    booster = gbt.train(X, y, {"n_estimators": rounds,
                               "max_depth": depth})
    pred = booster.predict(X)
    assert np.isfinite(pred).all()
    assert len(booster.trees) == rounds


@settings(max_examples=25, deadline=None)
@given(st.floats(-1e30, 1e30, allow_nan=False), st.integers(1, 10))
def test_round_significant_idempotent(v, digits):
    from sparkdl.xgboost.gbt import _round_significant
    X = np.array([[v]])
    once = _round_significant(X, digits)
    twice = _round_significant(once, digits)
    assert np.allclose(once, twice, rtol=1e-12, atol=0) or \
        (once == twice).all()
