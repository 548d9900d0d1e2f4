"""Histogram GBT family (CPU reference tier)."""

import numpy as np
import pytest
from sklearn.ensemble import HistGradientBoostingClassifier
from sklearn.metrics import r2_score

import spark_sklearn_amd as ska
from spark_sklearn_amd.models import (GradientBoostingClassifier,
                                      GradientBoostingRegressor)
from spark_sklearn_amd.models._gbt_ref import (GBTSpec, bin_data,
                                               gbt_fit_ref, gbt_predict_raw,
                                               quantile_bins)


@pytest.fixture(scope="module")
def problem():
    rng = np.random.default_rng(0)
    n, f = 3000, 16
    X = rng.standard_normal((n, f))
    y = ((X[:, 0] * X[:, 1] + X[:, 2] +
          0.3 * rng.standard_normal(n)) > 0).astype(np.int64)
    return X, y


def test_classifier_quality_vs_sklearn_hist(problem):
    X, y = problem
    ours = GradientBoostingClassifier(n_estimators=40, max_depth=3,
                                      learning_rate=0.2).fit(X, y)
    sk = HistGradientBoostingClassifier(max_iter=40, max_depth=3,
                                        learning_rate=0.2).fit(X, y)
    acc_ours = (ours.predict(X) == y).mean()
    assert acc_ours > sk.score(X, y) - 0.05
    proba = ours.predict_proba(X[:10])
    assert proba.shape == (10, 2)
    np.testing.assert_allclose(proba.sum(axis=1), 1.0)
    assert ours.n_iter_ == 40


def test_regressor_quality(problem):
    X, _ = problem
    rng = np.random.default_rng(1)
    yr = 2 * X[:, 0] + np.sin(X[:, 1]) + 0.1 * rng.standard_normal(len(X))
    r = GradientBoostingRegressor(n_estimators=60, max_depth=3).fit(X, yr)
    assert r2_score(yr, r.predict(X)) > 0.97


def test_monotone_loss_decrease(problem):
    """More trees never hurt training fit (boosting sanity)."""
    X, y = problem
    edges = quantile_bins(X)
    Xb = bin_data(X, edges)
    yf = y.astype(np.float32)
    accs = []
    for ne in (5, 20, 60):
        trees, f0 = gbt_fit_ref(Xb, yf, GBTSpec(n_estimators=ne,
                                                learning_rate=0.3))
        raw = gbt_predict_raw(Xb, trees, f0)
        accs.append(float(np.mean((raw > 0) == y)))
    assert accs[0] <= accs[1] + 1e-9 <= accs[2] + 2e-9


def test_min_samples_leaf_and_reg(problem):
    X, y = problem
    deep = GradientBoostingClassifier(n_estimators=20, max_depth=6,
                                      min_samples_leaf=1).fit(X, y)
    shallow = GradientBoostingClassifier(n_estimators=20, max_depth=6,
                                         min_samples_leaf=200).fit(X, y)
    # stronger leaf constraint regularizes -> lower train accuracy
    assert (deep.predict(X) == y).mean() >= \
        (shallow.predict(X) == y).mean() - 1e-9


def test_gbt_grid_search(problem):
    X, y = problem
    gs = ska.GridSearchCV(None,
                          GradientBoostingClassifier(n_estimators=20),
                          {"max_depth": [2, 4],
                           "learning_rate": [0.1, 0.3]}, cv=3,
                          return_train_score=False).fit(X, y)
    assert len(gs.cv_results_["params"]) == 4
    assert gs.best_score_ > 0.8
    assert gs.predict(X[:20]).shape == (20,)


def test_multiclass_softmax(problem):
    """K>2: one tree per class per round (softmax boosting)."""
    rng = np.random.default_rng(7)
    n, f, K = 2500, 10, 4
    centers = rng.standard_normal((K, f)) * 2.5
    y = rng.integers(0, K, n)
    X = (centers[y] + rng.standard_normal((n, f))).astype(np.float32)
    clf = GradientBoostingClassifier(n_estimators=20, max_depth=3,
                                     learning_rate=0.3).fit(X, y)
    assert (clf.predict(X) == y).mean() > 0.95
    proba = clf.predict_proba(X[:7])
    assert proba.shape == (7, K)
    np.testing.assert_allclose(proba.sum(axis=1), 1.0, rtol=1e-5)
    # string labels round-trip through classes_
    labels = np.array(["a", "b", "c", "d"])[y]
    clf2 = GradientBoostingClassifier(n_estimators=10, max_depth=3,
                                      learning_rate=0.3).fit(X, labels)
    assert set(clf2.predict(X[:20])) <= set("abcd")


def test_multiclass_grid_search(problem):
    rng = np.random.default_rng(3)
    n, f, K = 1500, 8, 3
    centers = rng.standard_normal((K, f)) * 2
    y = rng.integers(0, K, n)
    X = (centers[y] + rng.standard_normal((n, f))).astype(np.float32)
    gs = ska.GridSearchCV(None, GradientBoostingClassifier(n_estimators=10),
                          {"max_depth": [2, 3]}, cv=3,
                          return_train_score=False).fit(X, y)
    assert gs.best_score_ > 0.9
    assert gs.predict(X[:5]).shape == (5,)


def test_param_validation(problem):
    X, y = problem
    with pytest.raises(ValueError):
        GradientBoostingClassifier(max_depth=0).fit(X, y)
    with pytest.raises(ValueError):
        GradientBoostingClassifier(n_estimators=0).fit(X, y)


def test_feature_importances(problem):
    rng = np.random.default_rng(0)
    X = rng.standard_normal((2000, 8)).astype(np.float32)
    y = (X[:, 2] > 0).astype(np.int64)
    clf = GradientBoostingClassifier(n_estimators=10, max_depth=2).fit(X, y)
    fi = clf.feature_importances_
    assert fi.shape == (8,)
    assert abs(fi.sum() - 1.0) < 1e-9
    assert fi.argmax() == 2
