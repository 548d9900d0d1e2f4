"""CPU tests for the RandomForest family (reference semantics).

Device-vs-reference structure tests live in tests/test_gpu_rf.py.
"""

import numpy as np
import pytest

from sklearn.datasets import load_digits, make_classification, make_regression

from spark_sklearn_amd import GridSearchCV
from spark_sklearn_amd.models import (RandomForestClassifier,
                                      RandomForestRegressor)
from spark_sklearn_amd.models._rf_ref import (RFSpec, _bootstrap_positions,
                                              rf_fit_ref, rf_hash,
                                              rf_predict_ref)


def test_rf_hash_reference_vector():
    # pinned values so the HIP implementation can be checked bit-for-bit
    assert int(rf_hash(np.uint32(0))) == 0
    vals = [int(rf_hash(np.uint32(v))) for v in (1, 2, 12345, 0xDEADBEEF)]
    assert len(set(vals)) == 4
    assert all(0 <= v < 2 ** 32 for v in vals)
    # stable across calls
    assert vals == [int(rf_hash(np.uint32(v)))
                    for v in (1, 2, 12345, 0xDEADBEEF)]


def test_bootstrap_positions_uniform():
    pos = _bootstrap_positions(10_000, 42)
    assert pos.min() >= 0 and pos.max() < 10_000
    # with-replacement: about 1 - 1/e unique
    frac = len(np.unique(pos)) / 10_000
    assert 0.58 < frac < 0.68


def test_rf_binary_accuracy_vs_sklearn():
    X, y = make_classification(n_samples=3000, n_features=20,
                               n_informative=10, random_state=7)
    rf = RandomForestClassifier(n_estimators=30, max_depth=6,
                                random_state=0).fit(X, y)
    acc = (rf.predict(X) == y).mean()
    from sklearn.ensemble import RandomForestClassifier as SkRF

    sk = SkRF(n_estimators=30, max_depth=6, random_state=0).fit(X, y)
    sk_acc = (sk.predict(X) == y).mean()
    assert acc > 0.85
    assert abs(acc - sk_acc) < 0.08
    proba = rf.predict_proba(X)
    assert proba.shape == (len(y), 2)
    assert np.allclose(proba.sum(axis=1), 1.0, atol=1e-5)


def test_rf_multiclass_digits():
    X, y = load_digits(return_X_y=True)
    rf = RandomForestClassifier(n_estimators=25, max_depth=6,
                                random_state=3).fit(X, y)
    assert (rf.predict(X) == y).mean() > 0.93
    proba = rf.predict_proba(X)
    assert proba.shape == (len(y), 10)
    assert np.allclose(proba.sum(axis=1), 1.0, atol=1e-4)


def test_rf_regressor():
    X, y = make_regression(n_samples=2000, n_features=12, noise=5.0,
                           random_state=1)
    rf = RandomForestRegressor(n_estimators=25, max_depth=6,
                               random_state=0).fit(X, y)
    from sklearn.metrics import r2_score

    assert r2_score(y, rf.predict(X)) > 0.6


def test_rf_no_bootstrap_deterministic_data_fit():
    X, y = make_classification(n_samples=800, n_features=10,
                               random_state=5)
    a = RandomForestClassifier(n_estimators=5, max_depth=4, bootstrap=False,
                               max_features=None, random_state=0).fit(X, y)
    # all trees identical without bootstrap/feature subsampling
    f0, t0, v0 = a._trees[0]
    f1, t1, v1 = a._trees[1]
    assert np.array_equal(f0, f1) and np.array_equal(t0, t1)
    assert np.allclose(v0, v1)


def test_rf_feature_subsampling_varies_trees():
    X, y = make_classification(n_samples=800, n_features=20,
                               n_informative=10, random_state=5)
    a = RandomForestClassifier(n_estimators=6, max_depth=4, bootstrap=False,
                               max_features=4, random_state=0).fit(X, y)
    roots = {int(t[0][0]) for t in a._trees}
    assert len(roots) > 1      # different nodes saw different features


def test_rf_grid_search_digits_reference_workload():
    """The reference README's motivating shape: digits + RF grid."""
    X, y = load_digits(return_X_y=True)
    gs = GridSearchCV(None,
                      RandomForestClassifier(random_state=0),
                      {"max_depth": [6], "n_estimators": [10, 20]},
                      cv=3)
    gs.fit(X, y)
    assert gs.best_score_ > 0.85
    assert set(gs.best_params_) == {"max_depth", "n_estimators"}
    assert gs.cv_results_["mean_test_score"].shape == (2,)
    # refit model predicts
    assert (gs.best_estimator_.predict(X) == y).mean() > 0.9


def test_rf_max_depth_none_raises_clearly():
    X, y = make_classification(n_samples=100, random_state=0)
    rf = RandomForestClassifier(max_depth=None)
    with pytest.raises(ValueError, match="max_depth"):
        rf.fit(X, y)


def test_rf_error_score_contract():
    X, y = make_classification(n_samples=200, random_state=0)
    from sklearn.exceptions import FitFailedWarning

    gs = GridSearchCV(None, RandomForestClassifier(random_state=0),
                      {"max_depth": [3, 99]}, cv=2, error_score=0.0)
    with pytest.warns(FitFailedWarning):
        gs.fit(X, y)
    scores = gs.cv_results_["mean_test_score"]
    assert scores[1] == 0.0 and scores[0] > 0.5


def test_rf_ref_multiclass_planes_share_structure():
    rng = np.random.default_rng(0)
    Xb = rng.integers(0, 32, size=(500, 8), dtype=np.uint8)
    y = rng.integers(0, 4, size=500).astype(np.float32)
    spec = RFSpec(n_estimators=3, max_depth=3, n_classes=4, random_state=1)
    trees = rf_fit_ref(Xb, y, spec)
    for planes in trees:
        f0 = planes[0][0]
        for f, t, v in planes[1:]:
            assert np.array_equal(f, f0)
    raw = rf_predict_ref(Xb, trees, 4)
    assert raw.shape == (500, 4)
    # forest-average probabilities sum to ~1
    assert np.allclose(raw.sum(axis=1), 1.0, atol=1e-4)


def test_extratrees_classifier_and_regressor():
    X, y = make_classification(n_samples=1500, n_features=16,
                               n_informative=8, random_state=0)
    from spark_sklearn_amd.models import (ExtraTreesClassifier,
                                          ExtraTreesRegressor)

    et = ExtraTreesClassifier(n_estimators=20, max_depth=8,
                              random_state=0).fit(X, y)
    acc = (et.predict(X) == y).mean()
    from sklearn.ensemble import ExtraTreesClassifier as SkET

    sk = SkET(n_estimators=20, max_depth=8, random_state=0).fit(X, y)
    assert acc > 0.85
    assert abs(acc - (sk.predict(X) == y).mean()) < 0.08
    # no bootstrap by default, random thresholds still vary trees
    roots = {int(t[0][0]) for t in et._trees}
    assert len(roots) >= 1

    Xr, yr = make_regression(n_samples=1200, n_features=10, noise=3.0,
                             random_state=1)
    er = ExtraTreesRegressor(n_estimators=20, max_depth=8,
                             random_state=0).fit(Xr, yr)
    from sklearn.metrics import r2_score

    assert r2_score(yr, er.predict(Xr)) > 0.5


def test_extratrees_grid_search():
    # small shapes: the CPU tier runs the numpy reference forest
    X, y = make_classification(n_samples=600, n_features=12,
                               n_informative=6, random_state=2)
    from spark_sklearn_amd.models import ExtraTreesClassifier

    gs = GridSearchCV(None, ExtraTreesClassifier(random_state=0),
                      {"max_depth": [6], "n_estimators": [10, 20]}, cv=2)
    gs.fit(X, y)
    assert gs.best_score_ > 0.8
