"""class_weight for the batched classifiers.

Maps onto the sample_weight machinery: a dict weights rows by their
class, 'balanced' uses each TRAIN FOLD's class frequencies (sklearn
computes it on the data every fit sees).  Per-candidate class_weight
grids raise (one weight vector per kernel batch).
"""

import warnings

import numpy as np
import pytest

import spark_sklearn_amd as ska
from spark_sklearn_amd.models import (LogisticRegression,
                                      RandomForestClassifier)


def _imbalanced(n=1500, f=10, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    y = (X[:, 0] > 1.0).astype(int)          # ~16% positives
    return X, y


def test_dict_class_weight_equals_manual_sample_weight():
    X, y = _imbalanced()
    cw = {0: 1.0, 1: 4.0}
    m1 = LogisticRegression(max_iter=15, class_weight=cw).fit(X, y)
    w = np.where(y == 1, 4.0, 1.0)
    m2 = LogisticRegression(max_iter=15).fit(X, y, sample_weight=w)
    np.testing.assert_allclose(m1.coef_, m2.coef_, rtol=1e-6)
    np.testing.assert_allclose(m1.intercept_, m2.intercept_, rtol=1e-6)


def test_balanced_equals_manual_frequencies():
    X, y = _imbalanced(seed=3)
    n, K = len(y), 2
    cnt = np.bincount(y, minlength=K)
    w = (n / (K * cnt))[y]
    m1 = LogisticRegression(max_iter=15,
                            class_weight="balanced").fit(X, y)
    m2 = LogisticRegression(max_iter=15).fit(X, y, sample_weight=w)
    np.testing.assert_allclose(m1.coef_, m2.coef_, rtol=1e-6)


def test_balanced_raises_minority_recall():
    X, y = _imbalanced(seed=1)
    m0 = LogisticRegression(max_iter=25).fit(X, y)
    m1 = LogisticRegression(max_iter=25,
                            class_weight="balanced").fit(X, y)
    r0 = ((m0.predict(X) == 1) & (y == 1)).sum() / max((y == 1).sum(), 1)
    r1 = ((m1.predict(X) == 1) & (y == 1)).sum() / max((y == 1).sum(), 1)
    assert r1 > r0


def test_grid_search_balanced_per_fold():
    """'balanced' recomputes per train fold inside the search and
    changes the scores vs unweighted; no ignored-warnings fire."""
    X, y = _imbalanced(seed=2)
    gs = ska.GridSearchCV(
        None, LogisticRegression(max_iter=12, class_weight="balanced"),
        {"C": [0.1, 1.0]}, cv=3)
    with warnings.catch_warnings():
        warnings.filterwarnings("error", message=".*IGNORED.*")
        gs.fit(X, y)
    gs0 = ska.GridSearchCV(None, LogisticRegression(max_iter=12),
                           {"C": [0.1, 1.0]}, cv=3)
    gs0.fit(X, y)
    assert not np.allclose(gs.cv_results_["mean_test_score"],
                           gs0.cv_results_["mean_test_score"])


def test_uniform_candidate_class_weight_allowed():
    X, y = _imbalanced(seed=4)
    gs = ska.GridSearchCV(
        None, LogisticRegression(max_iter=10),
        {"C": [0.1, 1.0], "class_weight": [{0: 1.0, 1: 3.0}]}, cv=2)
    gs.fit(X, y)
    assert len(gs.cv_results_["params"]) == 2


def _engine_tasks(X, y, params_list, cv=2):
    from sklearn.model_selection import KFold

    from spark_sklearn_amd.base_search import Task

    splits = list(KFold(n_splits=cv).split(np.arange(len(y))))
    tasks, t = [], 0
    for ci, params in enumerate(params_list):
        for fi, (tr, te) in enumerate(splits):
            tasks.append(Task(t, ci, fi, params, tr, te))
            t += 1
    return tasks


def test_per_candidate_class_weight_grid_on_batched_path():
    """Per-candidate class_weight grids run on the batched path via
    task GROUPING (one weight vector per kernel batch): each group's
    scores equal the same candidate run alone."""
    from spark_sklearn_amd.engine import run_batched_tasks

    X, y = _imbalanced(seed=5)
    grid = [{"class_weight": None}, {"class_weight": "balanced"},
            {"class_weight": {0: 1.0, 1: 2.0}}]
    tasks = _engine_tasks(X, y, grid)
    res = run_batched_tasks(
        LogisticRegression(max_iter=10), X, y, tasks, scorer=None,
        fit_params=None, return_train_score=False,
        error_score="raise", verbose=0, device=None)
    by_idx = {r.index: r.test_score for r in res}
    assert len(by_idx) == len(tasks)

    for params in grid:
        solo = _engine_tasks(X, y, [params])
        res_s = run_batched_tasks(
            LogisticRegression(max_iter=10, **params), X, y, solo,
            scorer=None, fit_params=None, return_train_score=False,
            error_score="raise", verbose=0, device=None)
        # match fold-for-fold against the grouped run
        gi = grid.index(params)
        for fi, r in enumerate(res_s):
            grouped = by_idx[gi * 2 + fi]
            np.testing.assert_allclose(grouped, r.test_score, rtol=1e-9)

    # thread tier: the same grid also works (per-task fits)
    gs = ska.GridSearchCV(
        "threads", LogisticRegression(max_iter=10),
        {"class_weight": [None, "balanced"]}, cv=2)
    gs.fit(X, y)
    assert len(gs.cv_results_["params"]) == 2


def test_batched_path_balanced_matches_manual():
    """run_batched_tasks (host-executor tier) applies 'balanced' per
    train fold -- fold scores match the same search run with the
    per-fold manual weights."""
    from spark_sklearn_amd.engine import run_batched_tasks

    X, y = _imbalanced(seed=10)
    tasks = _engine_tasks(X, y, [{"C": 0.5}])
    res_b = run_batched_tasks(
        LogisticRegression(max_iter=10, class_weight="balanced"), X, y,
        tasks, scorer=None, fit_params=None, return_train_score=False,
        error_score="raise", verbose=0, device=None)
    scores_b = [r.test_score for r in res_b]

    scores_m = []
    for t in tasks:
        cnt = np.bincount(y[t.train], minlength=2)
        w = (len(t.train) / (2 * cnt))[y]
        res = run_batched_tasks(
            LogisticRegression(max_iter=10), X, y, [t], scorer=None,
            fit_params={"sample_weight": w}, return_train_score=False,
            error_score="raise", verbose=0, device=None)
        scores_m.append(res[0].test_score)
    np.testing.assert_allclose(scores_b, scores_m, rtol=1e-9)


def test_rf_class_weight_changes_forest():
    X, y = _imbalanced(n=800, f=8, seed=6)
    m0 = RandomForestClassifier(n_estimators=4, max_depth=3,
                                random_state=0).fit(X, y)
    m1 = RandomForestClassifier(n_estimators=4, max_depth=3,
                                random_state=0,
                                class_weight={0: 1.0, 1: 8.0}).fit(X, y)
    same = all(np.array_equal(a[0], b[0])
               for a, b in zip(m0._trees, m1._trees))
    assert not same

    # dict class_weight == the equivalent sample_weight, tree-exact
    w = np.where(y == 1, 8.0, 1.0)
    m2 = RandomForestClassifier(n_estimators=4, max_depth=3,
                                random_state=0).fit(X, y,
                                                    sample_weight=w)
    for (fa, ta, va), (fb, tb, vb) in zip(m1._trees, m2._trees):
        np.testing.assert_array_equal(fa, fb)
        np.testing.assert_allclose(va, vb, rtol=1e-5)


def test_class_weight_combines_with_sample_weight():
    X, y = _imbalanced(seed=7)
    cw = {0: 1.0, 1: 2.0}
    rng = np.random.default_rng(8)
    sw = rng.uniform(0.5, 2.0, size=len(y))
    m1 = LogisticRegression(max_iter=12, class_weight=cw).fit(
        X, y, sample_weight=sw)
    m2 = LogisticRegression(max_iter=12).fit(
        X, y, sample_weight=sw * np.where(y == 1, 2.0, 1.0))
    np.testing.assert_allclose(m1.coef_, m2.coef_, rtol=1e-5)


def test_bad_class_weight_values():
    X, y = _imbalanced(n=400, seed=9)
    with pytest.raises(ValueError, match="not a class label"):
        LogisticRegression(max_iter=5,
                           class_weight={7: 2.0}).fit(X, y)
    with pytest.raises(ValueError, match="'balanced' or a dict"):
        LogisticRegression(max_iter=5,
                           class_weight="weird").fit(X, y)
