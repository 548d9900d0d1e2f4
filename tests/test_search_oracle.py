"""Conformance tests: our search vs the installed sklearn oracle.

The reference's signature test move is running sklearn's own suite against
its Spark implementation; the modern equivalent
here is structural + numerical comparison of ``cv_results_`` against
``sklearn.model_selection.GridSearchCV`` on identical (dataset, grid, cv,
seed) inputs.
"""

import pickle

import numpy as np
import pytest
from sklearn import svm
from sklearn.model_selection import GridSearchCV as SKGridSearchCV
from sklearn.model_selection import KFold
from sklearn.model_selection import RandomizedSearchCV as SKRandomizedSearchCV
from sklearn.tree import DecisionTreeClassifier

import spark_sklearn_amd as ska

NUMERIC_KEYS = ["mean_test_score", "std_test_score", "mean_train_score",
                "std_train_score"]


def _compare_cv_results(ours, theirs, n_splits, check_times=True):
    for k in NUMERIC_KEYS:
        assert np.allclose(ours[k], theirs[k], atol=1e-12), k
    for i in range(n_splits):
        assert np.allclose(ours[f"split{i}_test_score"],
                           theirs[f"split{i}_test_score"]), f"split{i}"
    assert np.array_equal(ours["rank_test_score"], theirs["rank_test_score"])
    assert ours["params"] == list(theirs["params"])
    for k in theirs:
        if k.startswith("param_"):
            assert k in ours, k
            assert list(ours[k]) == list(theirs[k]), k
            assert list(ours[k].mask) == list(np.asarray(theirs[k].mask).ravel()) \
                if hasattr(theirs[k], "mask") else True
    if check_times:
        for k in ("mean_fit_time", "std_fit_time", "mean_score_time",
                  "std_score_time"):
            assert k in ours and len(ours[k]) == len(theirs[k])


@pytest.mark.parametrize("backend", ["threads", "sequential", "threads:2"])
def test_grid_search_matches_sklearn_svc(digits_std, backend):
    X, y = digits_std
    grid = {"C": [0.1, 1.0, 10.0], "kernel": ["linear", "rbf"]}
    cv = KFold(n_splits=3)
    ours = ska.GridSearchCV(backend, svm.SVC(random_state=0), grid, cv=cv,
                            return_train_score=True).fit(X, y)
    ref = SKGridSearchCV(svm.SVC(random_state=0), grid, cv=cv,
                         return_train_score=True).fit(X, y)
    _compare_cv_results(ours.cv_results_, ref.cv_results_, 3)
    assert ours.best_index_ == ref.best_index_
    assert ours.best_params_ == ref.best_params_
    assert np.isclose(ours.best_score_, ref.best_score_)
    assert ours.n_splits_ == ref.n_splits_
    # refit equivalence: same predictions
    assert np.array_equal(ours.predict(X), ref.predict(X))


def test_heterogeneous_grid_masked_params(digits_std):
    """List-of-dicts grids produce masked param_* arrays like sklearn."""
    X, y = digits_std
    grid = [{"kernel": ["linear"], "C": [0.5, 1.0]},
            {"kernel": ["rbf"], "C": [1.0], "gamma": [0.001, 0.01]}]
    cv = KFold(n_splits=3)
    ours = ska.GridSearchCV(None, svm.SVC(), grid, cv=cv).fit(X, y)
    ref = SKGridSearchCV(svm.SVC(), grid, cv=cv).fit(X, y)
    pg = ours.cv_results_["param_gamma"]
    pg_ref = ref.cv_results_["param_gamma"]
    assert list(np.asarray(pg.mask).ravel()) == \
        list(np.asarray(np.ma.getmaskarray(np.ma.asarray(pg_ref))).ravel())
    assert np.allclose(ours.cv_results_["mean_test_score"],
                       ref.cv_results_["mean_test_score"])
    assert np.array_equal(ours.cv_results_["rank_test_score"],
                          ref.cv_results_["rank_test_score"])


def test_randomized_search_matches_sklearn(digits_std):
    """Same random_state -> same sampled candidate set -> same results."""
    X, y = digits_std
    dists = {"max_depth": [2, 4, 6, 8, None],
             "min_samples_split": [2, 4, 8, 16],
             "criterion": ["gini", "entropy"]}
    cv = KFold(n_splits=3)
    est = DecisionTreeClassifier(random_state=7)
    ours = ska.RandomizedSearchCV(None, est, dists, n_iter=8, cv=cv,
                                  random_state=42,
                                  return_train_score=True).fit(X, y)
    ref = SKRandomizedSearchCV(est, dists, n_iter=8, cv=cv, random_state=42,
                               return_train_score=True).fit(X, y)
    assert ours.cv_results_["params"] == list(ref.cv_results_["params"])
    _compare_cv_results(ours.cv_results_, ref.cv_results_, 3)
    assert ours.best_index_ == ref.best_index_


def test_tie_ranking_first_of_tied_best():
    """rank via rankdata(-means,'min'); best_index_ = first of tied best."""
    import spark_sklearn_amd.base_search as bs

    class FakeResult:
        pass

    search = ska.GridSearchCV(None, svm.SVC(), {"C": [1.0]})
    results = []
    scores = [0.5, 0.9, 0.9, 0.2]
    idx = 0
    for cand in range(4):
        for split in range(2):
            results.append(bs.TaskResult(idx, scores[cand], scores[cand],
                                         10, 0.0, 0.0))
            idx += 1
    out = search._assemble_cv_results(
        [{"C": float(c)} for c in range(4)], results, 4, 2)
    assert list(out["rank_test_score"]) == [3, 1, 1, 4]
    best = int(np.flatnonzero(out["rank_test_score"] == 1)[0])
    assert best == 1  # first of the tied best


def test_scoring_string_and_iid(digits_std):
    X, y = digits_std
    cv = KFold(n_splits=3)
    grid = {"C": [0.5, 1.0]}
    ours = ska.GridSearchCV(None, svm.SVC(), grid, cv=cv,
                            scoring="accuracy").fit(X, y)
    ref = SKGridSearchCV(svm.SVC(), grid, cv=cv, scoring="accuracy").fit(X, y)
    assert np.allclose(ours.cv_results_["mean_test_score"],
                       ref.cv_results_["mean_test_score"])
    # iid=True applies n_test weighting (reference-era semantics)
    weighted = ska.GridSearchCV(None, svm.SVC(), grid, cv=cv,
                                iid=True).fit(X, y)
    assert weighted.cv_results_["mean_test_score"].shape == (2,)


def test_pickle_roundtrip(digits_std):
    X, y = digits_std
    gs = ska.GridSearchCV(None, svm.SVC(), {"C": [1.0]}, cv=3).fit(X, y)
    blob = pickle.dumps(gs)
    gs2 = pickle.loads(blob)
    assert np.array_equal(gs2.predict(X[:10]), gs.predict(X[:10]))
    assert gs2.best_params_ == gs.best_params_


def test_reference_readme_example(digits_std):
    """The reference README's signature workload: digits x RandomForest
    grid (scaled down) through our search, vs the sklearn oracle."""
    from sklearn.ensemble import RandomForestClassifier
    from sklearn.model_selection import GridSearchCV as SKGS

    X, y = digits_std
    grid = {"max_depth": [3, None], "n_estimators": [10, 20],
            "min_samples_split": [2, 10]}
    est = RandomForestClassifier(random_state=0)
    ours = ska.GridSearchCV(None, est, grid, cv=3,
                            return_train_score=False).fit(X, y)
    ref = SKGS(est, grid, cv=3, return_train_score=False).fit(X, y)
    np.testing.assert_allclose(ours.cv_results_["mean_test_score"],
                               ref.cv_results_["mean_test_score"],
                               rtol=1e-12)
    assert ours.best_params_ == ref.best_params_
    assert ours.predict(X[:10]).shape == (10,)
