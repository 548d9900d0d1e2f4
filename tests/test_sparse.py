"""Sparse CSR input path (BASELINE config 3 shape, CPU tier)."""

import numpy as np
import pytest
import scipy.sparse as sp

import spark_sklearn_amd as ska
from spark_sklearn_amd.models import LinearSVC, LogisticRegression, SGDClassifier


@pytest.fixture(scope="module")
def sparse_problem():
    rng = np.random.default_rng(1)
    n, f = 3000, 128
    mask = rng.random((n, f)) < 0.03
    Xd = rng.standard_normal((n, f)) * mask
    w = rng.standard_normal(f)
    y = (Xd @ w > 0).astype(np.int64)
    return sp.csr_matrix(Xd), Xd, y


def test_sparse_equals_dense_search(sparse_problem):
    Xs, Xd, y = sparse_problem
    grid = {"C": [0.01, 1.0, 100.0]}
    gs_s = ska.GridSearchCV(None, LogisticRegression(max_iter=60), grid,
                            cv=3).fit(Xs, y)
    gs_d = ska.GridSearchCV(None, LogisticRegression(max_iter=60), grid,
                            cv=3).fit(Xd, y)
    np.testing.assert_allclose(gs_s.cv_results_["mean_test_score"],
                               gs_d.cv_results_["mean_test_score"],
                               atol=1e-9)
    assert gs_s.best_params_ == gs_d.best_params_
    # refit predicts on sparse input
    np.testing.assert_array_equal(gs_s.predict(Xs[:100]),
                                  gs_d.predict(Xd[:100]))


def test_sparse_mixed_loss_grid(sparse_problem):
    """Config-3 shape: mixed hinge/log-loss candidates in one batch."""
    Xs, _, y = sparse_problem
    gs = ska.GridSearchCV(
        None, SGDClassifier(max_iter=60),
        {"alpha": [1e-5, 1e-3], "loss": ["hinge", "log_loss"]},
        cv=3).fit(Xs, y)
    assert len(gs.cv_results_["params"]) == 4
    assert gs.best_score_ > 0.8


def test_sparse_linearsvc(sparse_problem):
    Xs, _, y = sparse_problem
    est = LinearSVC(max_iter=80).fit(Xs, y)
    assert (est.predict(Xs) == y).mean() > 0.85


@pytest.mark.gpu
def test_sparse_device_ingest(sparse_problem):
    """CSR ingests into dense bf16 HBM residency; kernels run unchanged."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    Xs, Xd, y = sparse_problem
    gs = ska.GridSearchCV(None, LogisticRegression(max_iter=30, tol=0.0),
                          {"C": [0.1, 1.0]}, cv=3,
                          return_train_score=False).fit(Xs, y)
    gs_d = ska.GridSearchCV(None, LogisticRegression(max_iter=30, tol=0.0),
                            {"C": [0.1, 1.0]}, cv=3,
                            return_train_score=False).fit(Xd, y)
    np.testing.assert_allclose(gs.cv_results_["mean_test_score"],
                               gs_d.cv_results_["mean_test_score"],
                               atol=2e-3)
