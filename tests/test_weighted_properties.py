"""Property tests: integer sample_weight == row duplication.

For full-batch GD and histogram trees, a row with integer weight k is
algebraically identical to k copies of that row (given the engine's
mean-1 normalization on the weighted side, min_samples_leaf=1 so the
count-gated masks cannot diverge, and bootstrap off for RF).  Random
integer weights through hypothesis pin the weighted reference
implementations to the unweighted ones they generalize.
"""

import numpy as np
import pytest

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402


def _dup(X, y, w):
    reps = w.astype(int)
    return np.repeat(X, reps, axis=0), np.repeat(y, reps)


def _problem(seed, n=160, f=6):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    y = (X[:, 0] + 0.3 * rng.standard_normal(n) > 0).astype(np.float32)
    return X, y


@settings(max_examples=15, deadline=None)
@given(seed=st.integers(0, 10_000), loss=st.sampled_from([0, 1, 2]))
def test_linear_integer_weights_equal_duplication(seed, loss):
    from spark_sklearn_amd.models._solver_ref import (BatchPlan,
                                                      linear_fit_ref)

    X, y = _problem(seed)
    n = len(y)
    rng = np.random.default_rng(seed + 1)
    w = rng.integers(1, 4, size=n).astype(np.float64)
    X2, y2 = _dup(X, y, w)

    def plan(max_iter=8):
        B = 1
        return BatchPlan(
            loss=np.full(B, loss, np.int32),
            lr=np.full(B, 0.03, np.float32),
            l1=np.zeros(B, np.float32), l2=np.zeros(B, np.float32),
            momentum=np.full(B, 0.5, np.float32),
            tol=np.zeros(B, np.float32),
            max_iter=np.full(B, max_iter, np.int32),
            target_class=np.ones(B, np.int32),
            col_to_cand=np.zeros(B, np.int32), n_classes=1,
            fit_intercept=np.ones(B, np.uint8))

    w_norm = (w * (n / w.sum())).astype(np.float32)
    W_w, b_w, _, _ = linear_fit_ref(X, y, plan(), sample_weight=w_norm)
    W_d, b_d, _, _ = linear_fit_ref(X2, y2, plan())
    np.testing.assert_allclose(W_w[:, 0], W_d[:, 0], rtol=5e-4,
                               atol=5e-5)
    np.testing.assert_allclose(b_w, b_d, rtol=5e-4, atol=5e-5)


@settings(max_examples=10, deadline=None)
@given(seed=st.integers(0, 10_000))
def test_gbt_integer_weights_equal_duplication(seed):
    from spark_sklearn_amd.models._gbt_ref import (GBTSpec, bin_data,
                                                   gbt_fit_ref,
                                                   quantile_bins)

    X, y = _problem(seed, n=200)
    rng = np.random.default_rng(seed + 2)
    w = rng.integers(1, 4, size=len(y)).astype(np.float64)
    Xb = bin_data(X, quantile_bins(X))
    Xb2, y2 = _dup(Xb, y, w)
    spec = GBTSpec(loss=0, n_estimators=3, learning_rate=0.3,
                   max_depth=2, min_samples_leaf=1,
                   l2_regularization=0.0)
    tw, f0w = gbt_fit_ref(Xb, y, spec,
                          sample_weight=w.astype(np.float32))
    td, f0d = gbt_fit_ref(Xb2, y2, spec)
    assert abs(f0w - f0d) < 1e-5
    for (fa, ta, va), (fb, tb, vb) in zip(tw, td):
        np.testing.assert_array_equal(fa, fb)
        np.testing.assert_array_equal(ta, tb)
        np.testing.assert_allclose(va, vb, rtol=1e-3, atol=1e-5)


@settings(max_examples=10, deadline=None)
@given(seed=st.integers(0, 10_000), multiclass=st.booleans())
def test_rf_integer_weights_equal_duplication(seed, multiclass):
    from spark_sklearn_amd.models._gbt_ref import bin_data, quantile_bins
    from spark_sklearn_amd.models._rf_ref import RFSpec, rf_fit_ref

    X, y = _problem(seed, n=200)
    if multiclass:
        y = (np.digitize(X[:, 1], [-0.5, 0.5])).astype(np.float32)
    rng = np.random.default_rng(seed + 3)
    w = rng.integers(1, 4, size=len(y)).astype(np.float64)
    Xb = bin_data(X, quantile_bins(X))
    Xb2, y2 = _dup(Xb, y, w)
    K = 3 if multiclass else 0
    spec = RFSpec(n_estimators=2, max_depth=3, min_samples_leaf=1,
                  max_features=4, bootstrap=False, random_state=seed,
                  n_classes=K)
    tw = rf_fit_ref(Xb, y, spec, sample_weight=w)
    td = rf_fit_ref(Xb2, y2, spec)
    for pw, pd in zip(tw, td):
        aw = pw if multiclass else [pw]
        ad = pd if multiclass else [pd]
        for (fa, ta, va), (fb, tb, vb) in zip(aw, ad):
            np.testing.assert_array_equal(fa, fb)
            np.testing.assert_array_equal(ta, tb)
            np.testing.assert_allclose(va, vb, rtol=1e-3, atol=1e-5)
