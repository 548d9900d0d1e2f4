"""Successive-halving pruning (BASELINE config 5 extension)."""

import numpy as np
import pytest
from scipy.stats import loguniform

import spark_sklearn_amd as ska
from spark_sklearn_amd.models import LogisticRegression


@pytest.fixture(scope="module")
def problem():
    rng = np.random.default_rng(0)
    n, f = 1500, 24
    X = rng.standard_normal((n, f))
    w = rng.standard_normal(f)
    y = ((X @ w + 0.3 * rng.standard_normal(n)) > 0).astype(np.int64)
    return X, y


def test_halving_prunes_and_finds_good_candidate(problem):
    X, y = problem
    rs = ska.RandomizedSearchCV(
        None, LogisticRegression(max_iter=80, tol=0.0),
        {"C": loguniform(1e-4, 1e3)}, n_iter=24, cv=3, random_state=0,
        halving_factor=3, min_resource_frac=0.1,
        return_train_score=False)
    rs.fit(X, y)
    res = rs.cv_results_
    assert len(res["params"]) == 24
    assert "rung" in res and "resource_iter" in res
    # rungs shrink the population: only a few reach the full budget
    n_final = int(np.sum(res["resource_iter"] == 80))
    assert 1 <= n_final <= 24 // 3 + 2
    assert res["rung"].max() >= 2
    # best candidate comes from the final rung and is genuinely good
    assert res["resource_iter"][rs.best_index_] == 80
    assert rs.best_score_ > 0.85
    assert rs.predict(X).shape == (len(y),)


def test_halving_matches_full_search_quality(problem):
    """Pruned search should find a near-best candidate vs the full sweep."""
    X, y = problem
    common = dict(n_iter=16, cv=3, random_state=7,
                  return_train_score=False)
    full = ska.RandomizedSearchCV(
        None, LogisticRegression(max_iter=60, tol=0.0),
        {"C": loguniform(1e-4, 1e3)}, **common).fit(X, y)
    pruned = ska.RandomizedSearchCV(
        None, LogisticRegression(max_iter=60, tol=0.0),
        {"C": loguniform(1e-4, 1e3)}, halving_factor=2,
        min_resource_frac=0.15, **common).fit(X, y)
    # same candidate set (same seed), near-equal best quality
    assert pruned.cv_results_["params"] == full.cv_results_["params"]
    assert pruned.best_score_ >= full.best_score_ - 0.02


def test_halving_validation(problem):
    X, y = problem
    rs = ska.RandomizedSearchCV(
        None, LogisticRegression(), {"C": [1.0]}, n_iter=1,
        halving_factor=1.0)
    with pytest.raises(ValueError, match="halving_factor"):
        rs.fit(X, y)


def test_halving_clamps_grid_max_iter(problem):
    """ADVICE round-1: a grid that sweeps max_iter must not override the
    rung budget, and the schedule must reach the grid's largest value."""
    X, y = problem
    rs = ska.RandomizedSearchCV(
        None, LogisticRegression(max_iter=10, tol=0.0),
        {"C": loguniform(1e-3, 1e2), "max_iter": [5, 60]},
        n_iter=12, cv=2, random_state=0, halving_factor=3,
        min_resource_frac=0.1, return_train_score=False)
    rs.fit(X, y)
    res = rs.cv_results_
    # the final rung reached the largest requested budget
    assert res["resource_iter"].max() == 60
    # early rungs ran at a fraction of it
    assert res["resource_iter"].min() < 60
    assert rs.best_score_ > 0.8
