"""sample_weight routing for the linear family (ROADMAP #8).

The engine honors fit(..., sample_weight=w) for linear estimators on
both tiers: w is validated, normalized to mean 1, and scales each row's
loss AND gradient (reference: _solver_ref.linear_fit_ref sample_weight;
device: fwd_d3/fwd_d2 WT=1 epilogue).  Other families warn-and-ignore.
"""

import warnings

import numpy as np
import pytest

import spark_sklearn_amd as ska
from spark_sklearn_amd.models import KMeans, LogisticRegression
from spark_sklearn_amd.models._solver_ref import (
    BatchPlan, LOSS_LOGISTIC, linear_fit_ref)


def _data(n=4000, f=32, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    y = (X[:, 0] + 0.3 * rng.standard_normal(n) > 0).astype(int)
    return X, y


def test_unit_weights_equal_unweighted_exactly():
    X, y = _data()
    m0 = LogisticRegression(max_iter=20).fit(X, y)
    m1 = LogisticRegression(max_iter=20).fit(
        X, y, sample_weight=np.ones(len(y)))
    np.testing.assert_array_equal(m0.coef_, m1.coef_)
    np.testing.assert_array_equal(m0.intercept_, m1.intercept_)


def test_upweighting_moves_the_boundary():
    X, y = _data()
    w = np.where(y == 1, 10.0, 1.0)
    m0 = LogisticRegression(max_iter=20).fit(X, y)
    m2 = LogisticRegression(max_iter=20).fit(X, y, sample_weight=w)
    assert not np.allclose(m0.coef_, m2.coef_)
    rec0 = ((m0.predict(X) == 1) & (y == 1)).sum() / (y == 1).sum()
    rec2 = ((m2.predict(X) == 1) & (y == 1)).sum() / (y == 1).sum()
    assert rec2 > rec0  # minority-style upweighting raises class-1 recall


def test_weight_two_equals_duplicated_rows():
    """w=2 on a subset == duplicating those rows (full-batch GD with an
    explicit lr is exactly linear in per-row weight)."""
    X, y = _data(n=600, f=8, seed=3)
    dup = np.arange(0, 600, 3)
    X2 = np.vstack([X, X[dup]])
    y2 = np.concatenate([y, y[dup]])
    w = np.ones(600)
    w[dup] = 2.0

    def plan(n_rows):
        B = 1
        return BatchPlan(
            loss=np.full(B, LOSS_LOGISTIC, np.int32),
            lr=np.full(B, 0.05, np.float32),
            l1=np.zeros(B, np.float32), l2=np.zeros(B, np.float32),
            momentum=np.zeros(B, np.float32),
            tol=np.zeros(B, np.float32),
            max_iter=np.full(B, 10, np.int32),
            target_class=np.ones(B, np.int32),
            col_to_cand=np.zeros(B, np.int32), n_classes=1,
            fit_intercept=np.ones(B, np.uint8))

    # engine normalization: mean-1 over the weighted set
    w_norm = w * (len(w) / w.sum())
    W_w, b_w, _, _ = linear_fit_ref(X, y, plan(600),
                                    sample_weight=w_norm)
    W_d, b_d, _, _ = linear_fit_ref(X2, y2, plan(800))
    # identical up to fp32 summation-order noise
    np.testing.assert_allclose(W_w[:, 0], W_d[:, 0], rtol=2e-4,
                               atol=2e-5)
    np.testing.assert_allclose(b_w, b_d, rtol=2e-4, atol=2e-5)


def test_grid_search_routes_sample_weight_without_ignore_warning():
    X, y = _data(n=1200, f=16)
    w = np.where(y == 1, 4.0, 1.0)
    gs = ska.GridSearchCV(None, LogisticRegression(max_iter=10),
                          {"C": [0.1, 1.0]}, cv=3)
    with warnings.catch_warnings():
        warnings.filterwarnings("error", message=".*IGNORED.*")
        gs.fit(X, y, sample_weight=w)
    gs0 = ska.GridSearchCV(None, LogisticRegression(max_iter=10),
                           {"C": [0.1, 1.0]}, cv=3)
    gs0.fit(X, y)
    assert not np.allclose(gs.cv_results_["mean_test_score"],
                           gs0.cv_results_["mean_test_score"])


def test_gbt_sample_weight_reference_semantics():
    """GBT honors sample_weight: w=1 reproduces the unweighted model
    exactly, w=2 equals duplicated rows (histogram sums are linear in
    w), and upweighting a class moves predictions toward it."""
    from spark_sklearn_amd.models._gbt_ref import (
        GBTSpec, bin_data, gbt_fit_ref, gbt_predict_raw, quantile_bins)

    X, y = _data(n=900, f=8, seed=5)
    edges = quantile_bins(X)
    Xb = bin_data(X, edges)
    spec = GBTSpec(loss=0, n_estimators=5, learning_rate=0.3, max_depth=3,
                   min_samples_leaf=1, l2_regularization=0.0)
    yf = y.astype(np.float32)
    t0, f00 = gbt_fit_ref(Xb, yf, spec)
    t1, f01 = gbt_fit_ref(Xb, yf, spec, sample_weight=np.ones(900,
                                                             np.float32))
    assert f00 == f01
    for (fa, ta, va), (fb, tb, vb) in zip(t0, t1):
        np.testing.assert_array_equal(fa, fb)
        np.testing.assert_array_equal(ta, tb)
        np.testing.assert_allclose(va, vb)

    # w=2 on a subset == duplicating those rows (min_samples_leaf=1 so
    # the count channel cannot diverge the split sets)
    dup = np.arange(0, 900, 3)
    Xb2 = np.vstack([Xb, Xb[dup]])
    y2 = np.concatenate([yf, yf[dup]])
    w = np.ones(900, np.float32)
    w[dup] = 2.0
    tw, f0w = gbt_fit_ref(Xb, yf, spec, sample_weight=w)
    td, f0d = gbt_fit_ref(Xb2, y2, spec)
    assert abs(f0w - f0d) < 1e-6
    for (fa, ta, va), (fb, tb, vb) in zip(tw, td):
        np.testing.assert_array_equal(fa, fb)
        np.testing.assert_array_equal(ta, tb)
        np.testing.assert_allclose(va, vb, rtol=1e-4, atol=1e-6)

    # upweighting class 1 moves raw scores up
    wup = np.where(y == 1, 8.0, 1.0).astype(np.float32)
    tu, f0u = gbt_fit_ref(Xb, yf, spec, sample_weight=wup)
    r_base = gbt_predict_raw(Xb, t0, f00)
    r_up = gbt_predict_raw(Xb, tu, f0u)
    assert r_up.mean() > r_base.mean()


def test_gbt_grid_routes_sample_weight():
    from spark_sklearn_amd.models import GradientBoostingClassifier

    X, y = _data(n=800, f=8, seed=2)
    w = np.where(y == 1, 5.0, 1.0)
    gs = ska.GridSearchCV(
        None, GradientBoostingClassifier(n_estimators=5, max_depth=2),
        {"learning_rate": [0.1, 0.3]}, cv=2)
    with warnings.catch_warnings():
        warnings.filterwarnings("error", message=".*IGNORED.*")
        gs.fit(X, y, sample_weight=w)
    gs0 = ska.GridSearchCV(
        None, GradientBoostingClassifier(n_estimators=5, max_depth=2),
        {"learning_rate": [0.1, 0.3]}, cv=2)
    gs0.fit(X, y)
    assert not np.allclose(gs.cv_results_["mean_test_score"],
                           gs0.cv_results_["mean_test_score"])


def test_rf_sample_weight_reference_semantics():
    """RF honors sample_weight: weighted gini via w-scaled class planes
    + a sum(w) mass channel.  w=1 == unweighted exactly; w=2 ==
    duplicated rows (bootstrap=False, min_samples_leaf=1 so the
    count-based leaf gate cannot diverge)."""
    from spark_sklearn_amd.models._gbt_ref import bin_data, quantile_bins
    from spark_sklearn_amd.models._rf_ref import RFSpec, rf_fit_ref

    X, y = _data(n=600, f=8, seed=7)
    Xb = bin_data(X, quantile_bins(X))
    yf = y.astype(np.float32)
    spec = RFSpec(n_estimators=4, max_depth=3, min_samples_leaf=1,
                  max_features=4, bootstrap=False, random_state=0,
                  n_classes=0)
    t0 = rf_fit_ref(Xb, yf, spec)
    t1 = rf_fit_ref(Xb, yf, spec, sample_weight=np.ones(600))
    for (fa, ta, va), (fb, tb, vb) in zip(t0, t1):
        np.testing.assert_array_equal(fa, fb)
        np.testing.assert_array_equal(ta, tb)
        np.testing.assert_allclose(va, vb)

    dup = np.arange(0, 600, 3)
    Xb2 = np.vstack([Xb, Xb[dup]])
    y2 = np.concatenate([yf, yf[dup]])
    w = np.ones(600)
    w[dup] = 2.0
    tw = rf_fit_ref(Xb, yf, spec, sample_weight=w)
    td = rf_fit_ref(Xb2, y2, spec)
    for (fa, ta, va), (fb, tb, vb) in zip(tw, td):
        np.testing.assert_array_equal(fa, fb)
        np.testing.assert_array_equal(ta, tb)
        np.testing.assert_allclose(va, vb, rtol=1e-5, atol=1e-7)


def test_rf_grid_routes_sample_weight():
    from spark_sklearn_amd.models import RandomForestClassifier

    X, y = _data(n=700, f=8, seed=8)
    w = np.where(y == 1, 6.0, 1.0)
    gs = ska.GridSearchCV(
        None, RandomForestClassifier(n_estimators=4, max_depth=3,
                                     random_state=0),
        {"max_features": [2, 4]}, cv=2)
    with warnings.catch_warnings():
        warnings.filterwarnings("error", message=".*IGNORED.*")
        gs.fit(X, y, sample_weight=w)
    gs0 = ska.GridSearchCV(
        None, RandomForestClassifier(n_estimators=4, max_depth=3,
                                     random_state=0),
        {"max_features": [2, 4]}, cv=2)
    gs0.fit(X, y)
    assert not np.allclose(gs.cv_results_["mean_test_score"],
                           gs0.cv_results_["mean_test_score"])


def test_kmeans_sample_weight_weighted_centroids():
    """KMeans honors sample_weight: w=1 == unweighted exactly; a heavy
    cluster pulls its centroid toward the heavy points (weighted-mean
    update, reference tier)."""
    rng = np.random.default_rng(6)
    base = np.concatenate([rng.normal(-3, 0.3, (200, 2)),
                           rng.normal(+3, 0.3, (200, 2))]).astype(np.float32)
    m0 = KMeans(n_clusters=2, max_iter=20, random_state=0).fit(base)
    m1 = KMeans(n_clusters=2, max_iter=20, random_state=0).fit(
        base, sample_weight=np.ones(400))
    np.testing.assert_array_equal(m0.cluster_centers_, m1.cluster_centers_)

    # upweight the 20 right-most points of the right blob: that
    # centroid's x moves right
    w = np.ones(400)
    right = np.argsort(base[:, 0])[-20:]
    w[right] = 50.0
    m2 = KMeans(n_clusters=2, max_iter=20, random_state=0).fit(
        base, sample_weight=w)
    cx0 = sorted(m0.cluster_centers_[:, 0])
    cx2 = sorted(m2.cluster_centers_[:, 0])
    assert cx2[1] > cx0[1] + 0.1

    # grid routing: no ignored-warning
    gs = ska.GridSearchCV(None, KMeans(n_clusters=2, max_iter=5,
                                       random_state=0),
                          {"n_clusters": [2, 3]}, cv=2)
    with warnings.catch_warnings():
        warnings.filterwarnings("error", message=".*IGNORED.*")
        gs.fit(base, sample_weight=w)


def test_halving_sweep_routes_sample_weight():
    """Successive-halving (RandomizedSearchCV + halving_factor) carries
    fit_params into every rung."""
    from scipy.stats import loguniform

    X, y = _data(n=900, f=12, seed=4)
    w = np.where(y == 1, 5.0, 1.0)
    rs = ska.RandomizedSearchCV(
        None, LogisticRegression(max_iter=16),
        {"C": loguniform(1e-3, 1e3)}, n_iter=8, cv=2, random_state=0,
        halving_factor=2)
    with warnings.catch_warnings():
        warnings.filterwarnings("error", message=".*IGNORED.*")
        rs.fit(X, y, sample_weight=w)
    rs0 = ska.RandomizedSearchCV(
        None, LogisticRegression(max_iter=16),
        {"C": loguniform(1e-3, 1e3)}, n_iter=8, cv=2, random_state=0,
        halving_factor=2)
    rs0.fit(X, y)
    assert not np.allclose(
        np.nan_to_num(rs.cv_results_["mean_test_score"], nan=-1),
        np.nan_to_num(rs0.cv_results_["mean_test_score"], nan=-1))


def test_sample_weight_scale_invariance():
    """The engine normalizes w to mean 1, so a global rescale of the
    weights must not change any family's fitted model."""
    from spark_sklearn_amd.models import (
        GradientBoostingClassifier, RandomForestClassifier)

    X, y = _data(n=700, f=8, seed=9)
    w = np.where(y == 1, 3.0, 1.0)

    m1 = LogisticRegression(max_iter=15).fit(X, y, sample_weight=w)
    m2 = LogisticRegression(max_iter=15).fit(X, y, sample_weight=7.5 * w)
    np.testing.assert_allclose(m1.coef_, m2.coef_, rtol=1e-5)

    g1 = GradientBoostingClassifier(n_estimators=4, max_depth=2).fit(
        X, y, sample_weight=w)
    g2 = GradientBoostingClassifier(n_estimators=4, max_depth=2).fit(
        X, y, sample_weight=7.5 * w)
    for (fa, ta, va), (fb, tb, vb) in zip(g1._trees, g2._trees):
        np.testing.assert_array_equal(fa, fb)
        np.testing.assert_allclose(va, vb, rtol=1e-4, atol=1e-6)

    r1 = RandomForestClassifier(n_estimators=3, max_depth=3,
                                random_state=0).fit(X, y, sample_weight=w)
    r2 = RandomForestClassifier(n_estimators=3, max_depth=3,
                                random_state=0).fit(X, y,
                                                    sample_weight=7.5 * w)
    for (fa, ta, va), (fb, tb, vb) in zip(r1._trees, r2._trees):
        np.testing.assert_array_equal(fa, fb)
        np.testing.assert_allclose(va, vb, rtol=1e-4, atol=1e-6)


def test_sample_weight_validation():
    X, y = _data(n=500, f=8)
    with pytest.raises(ValueError, match="entries"):
        LogisticRegression(max_iter=5).fit(X, y,
                                           sample_weight=np.ones(3))
    with pytest.raises(ValueError, match="non-negative"):
        LogisticRegression(max_iter=5).fit(
            X, y, sample_weight=np.full(500, -1.0))
    with pytest.raises(ValueError, match="zero"):
        LogisticRegression(max_iter=5).fit(
            X, y, sample_weight=np.zeros(500))
