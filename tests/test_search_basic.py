"""Direct behavior tests of the search layer (error_score, refit,
delegation, native-family batched CPU path)."""

import numpy as np
import pytest
from sklearn import svm
from sklearn.base import BaseEstimator, ClassifierMixin

import spark_sklearn_amd as ska
from spark_sklearn_amd.models import (LinearSVC, LogisticRegression, Ridge,
                                      SGDClassifier)


class FailingEstimator(ClassifierMixin, BaseEstimator):
    """Raises during fit for C < 0 (error_score contract probe)."""

    def __init__(self, C=1.0):
        self.C = C

    def fit(self, X, y):
        if self.C < 0:
            raise ValueError("bad C")
        self.classes_ = np.unique(y)
        self.majority_ = np.bincount(y).argmax()
        return self

    def predict(self, X):
        return np.full(len(X), self.majority_)


def _xy(n=200, f=8, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f))
    y = (X[:, 0] + 0.1 * rng.standard_normal(n) > 0).astype(np.int64)
    return X, y


def test_error_score_numeric_swallows():
    X, y = _xy()
    from sklearn.exceptions import FitFailedWarning

    gs = ska.GridSearchCV(None, FailingEstimator(), {"C": [-1.0, 1.0]},
                          cv=3, error_score=-999.0, refit=True)
    with pytest.warns(FitFailedWarning, match="3 fits failed out of"):
        gs.fit(X, y)
    assert gs.cv_results_["mean_test_score"][0] == -999.0
    assert gs.best_params_ == {"C": 1.0}


def test_error_score_default_nan():
    """Default error_score matches the installed sklearn oracle (np.nan
    with an aggregated FitFailedWarning); 'raise' remains available."""
    X, y = _xy()
    from sklearn.exceptions import FitFailedWarning

    gs = ska.GridSearchCV(None, FailingEstimator(), {"C": [-1.0, 1.0]},
                          cv=3)
    with pytest.warns(FitFailedWarning):
        gs.fit(X, y)
    assert np.isnan(gs.cv_results_["mean_test_score"][0])

    gs2 = ska.GridSearchCV(None, FailingEstimator(), {"C": [-1.0, 1.0]},
                           cv=3, error_score="raise")
    with pytest.raises(ValueError, match="bad C"):
        gs2.fit(X, y)


def test_refit_false_blocks_predict():
    X, y = _xy()
    gs = ska.GridSearchCV(None, svm.SVC(), {"C": [1.0]}, cv=3, refit=False)
    gs.fit(X, y)
    assert not hasattr(gs, "best_estimator_")
    with pytest.raises(AttributeError):
        gs.predict(X)
    assert "mean_test_score" in gs.cv_results_


def test_param_grid_validation():
    X, y = _xy()
    with pytest.raises(TypeError):
        ska.GridSearchCV(None, svm.SVC(), {"C": "not-a-sequence"}).fit(X, y)
    with pytest.raises(ValueError):
        ska.GridSearchCV(None, svm.SVC(), {"C": []}).fit(X, y)


def test_verbose_and_groups(capsys):
    X, y = _xy()
    gs = ska.GridSearchCV(None, svm.SVC(), {"C": [1.0]}, cv=3, verbose=1)
    gs.fit(X, y)
    assert "Fitting 3 folds" in capsys.readouterr().out


def test_context_object():
    X, y = _xy()
    ctx = ska.createLocalContext(candidate_batch=64)
    gs = ska.GridSearchCV(ctx, svm.SVC(), {"C": [1.0, 2.0]}, cv=3).fit(X, y)
    assert gs.best_params_["C"] in (1.0, 2.0)
    assert "HipContext" in repr(ctx)


# ---- native batched family through the full search (CPU tier) -----------

@pytest.mark.parametrize("est,grid", [
    (LogisticRegression(max_iter=50), {"C": [0.1, 1.0, 10.0]}),
    (LinearSVC(max_iter=50), {"C": [0.1, 1.0]}),
    (SGDClassifier(max_iter=50), {"alpha": [1e-4, 1e-2],
                                  "loss": ["hinge", "log_loss"]}),
])
def test_native_family_grid_search(est, grid):
    X, y = _xy(400, 16)
    gs = ska.GridSearchCV(None, est, grid, cv=3).fit(X, y)
    n_cand = int(np.prod([len(v) for v in grid.values()]))
    assert len(gs.cv_results_["params"]) == n_cand
    assert gs.best_score_ > 0.8          # separable problem
    pred = gs.predict(X)
    assert pred.shape == (len(y),)
    assert (pred == y).mean() > 0.8


def test_native_multiclass_ovr(digits_std):
    X, y = digits_std
    gs = ska.GridSearchCV(None, LogisticRegression(max_iter=40),
                          {"C": [0.1, 1.0]}, cv=3).fit(X, y)
    assert gs.best_score_ > 0.85
    assert gs.best_estimator_.coef_.shape == (10, X.shape[1])
    proba = gs.predict_proba(X[:5])
    assert proba.shape == (5, 10)
    assert np.allclose(proba.sum(axis=1), 1.0, atol=1e-6)


def test_native_regression_family():
    rng = np.random.default_rng(3)
    X = rng.standard_normal((500, 12))
    w = rng.standard_normal(12)
    y = X @ w + 0.05 * rng.standard_normal(500)
    gs = ska.GridSearchCV(None, Ridge(max_iter=200),
                          {"alpha": [1e-3, 1.0, 100.0]}, cv=3).fit(X, y)
    assert gs.best_score_ > 0.95          # r2
    assert gs.best_params_["alpha"] < 100.0


def test_native_custom_scorer_fallback():
    """Non-device-computable scorer routes through host scoring."""
    X, y = _xy(300, 8)
    gs = ska.GridSearchCV(None, LogisticRegression(max_iter=40),
                          {"C": [0.1, 1.0]}, cv=3,
                          scoring="f1").fit(X, y)
    assert 0.0 <= gs.best_score_ <= 1.0
    assert gs.best_score_ > 0.7


def test_engine_multimetric_device_metric_loop():
    """run_batched_tasks with a _MultiScorer whose metrics are ALL
    device-computable scores each metric through the executor (host tier
    here; the identical code path runs the HIP executor on GPU)."""
    from sklearn.metrics import check_scoring
    from sklearn.model_selection import KFold

    from spark_sklearn_amd.base_search import Task, _MultiScorer
    from spark_sklearn_amd.engine import run_batched_tasks
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    X = rng.standard_normal((400, 8)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.int64)
    est = LogisticRegression(max_iter=25)
    scorers = {"acc": check_scoring(est, scoring="accuracy"),
               "nll": check_scoring(est, scoring="neg_log_loss")}
    ms = _MultiScorer(scorers)
    splits = list(KFold(3).split(np.arange(len(X))))
    tasks = []
    i = 0
    for ci, params in enumerate([{"C": 0.1}, {"C": 1.0}]):
        for fi, (tr, te) in enumerate(splits):
            tasks.append(Task(i, ci, fi, params, tr, te))
            i += 1
    results = run_batched_tasks(
        est, X, y, tasks, scorer=ms, fit_params=None,
        return_train_score=True, error_score="raise", verbose=0,
        device=None)
    assert len(results) == 6
    for r in results:
        assert set(r.test_score) == {"acc", "nll"}
        assert set(r.train_score) == {"acc", "nll"}
        assert 0.5 < r.test_score["acc"] <= 1.0
        assert r.test_score["nll"] <= 0.0
