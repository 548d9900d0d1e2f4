"""Numerics of the reference batched solver (the spec the HIP kernels match)."""

import numpy as np
import pytest
from sklearn.linear_model import LogisticRegression as SKLogReg
from sklearn.svm import LinearSVC as SKLinearSVC

from spark_sklearn_amd.models._solver_ref import (
    LOSS_HINGE, LOSS_LOGISTIC, LOSS_SQUARED, BatchPlan, auto_lr,
    linear_fit_ref, linear_scores_ref)


def _plan(B, loss, lr, l2=0.0, l1=0.0, max_iter=100, tol=0.0, mom=0.9,
          target=None, col_to_cand=None, n_classes=1, intercept=1):
    mk = lambda v, dt: np.full(B, v, dtype=dt)  # noqa: E731
    return BatchPlan(
        loss=mk(loss, np.int32), lr=np.asarray(lr, np.float32) * np.ones(B, np.float32),
        l1=mk(l1, np.float32), l2=np.asarray(l2, np.float32) * np.ones(B, np.float32),
        momentum=mk(mom, np.float32), tol=mk(tol, np.float32),
        max_iter=mk(max_iter, np.int32),
        target_class=(np.asarray(target, np.int32) if target is not None
                      else mk(1, np.int32)),
        col_to_cand=(np.asarray(col_to_cand, np.int32)
                     if col_to_cand is not None
                     else np.arange(B, dtype=np.int32)),
        n_classes=n_classes, fit_intercept=mk(intercept, np.uint8))


def test_logistic_converges_and_scores(binary_blobs):
    X, y = binary_blobs
    msq = float(np.mean(np.sum(X ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    plan = _plan(1, LOSS_LOGISTIC, lr, max_iter=200)
    W, b, n_iter, loss = linear_fit_ref(X, y, plan)
    acc = linear_scores_ref(X, y, W, b, plan, "accuracy")
    assert acc[0] > 0.95
    # comparable to sklearn (scores, not weights -- SURVEY.md 7.3)
    sk = SKLogReg(C=1e6, max_iter=500).fit(X, y)
    assert abs(acc[0] - sk.score(X, y)) < 0.02


def test_batched_columns_independent(binary_blobs):
    """Each column evolves independently: batch == per-column runs."""
    X, y = binary_blobs
    msq = float(np.mean(np.sum(X ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    l2s = np.array([0.0, 0.01, 1.0], dtype=np.float32)
    plan = _plan(3, LOSS_LOGISTIC, lr, l2=l2s, max_iter=50)
    W, b, _, _ = linear_fit_ref(X, y, plan)
    for j, l2 in enumerate(l2s):
        p1 = _plan(1, LOSS_LOGISTIC, lr, l2=float(l2), max_iter=50)
        W1, b1, _, _ = linear_fit_ref(X, y, p1)
        np.testing.assert_allclose(W[:, j], W1[:, 0], rtol=1e-5, atol=1e-6)
        np.testing.assert_allclose(b[j], b1[0], rtol=1e-5, atol=1e-6)


def test_hinge_comparable_to_linearsvc(binary_blobs):
    X, y = binary_blobs
    msq = float(np.mean(np.sum(X ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_HINGE, 0.0)
    plan = _plan(1, LOSS_HINGE, lr, l2=1.0 / len(y), max_iter=300)
    W, b, _, _ = linear_fit_ref(X, y, plan)
    acc = linear_scores_ref(X, y, W, b, plan, "accuracy")
    sk = SKLinearSVC(C=1.0).fit(X, y)
    assert abs(acc[0] - sk.score(X, y)) < 0.02


def test_squared_loss_regression():
    rng = np.random.default_rng(1)
    X = rng.standard_normal((1000, 16)).astype(np.float32)
    w = rng.standard_normal(16).astype(np.float32)
    y = X @ w + 2.5
    msq = float(np.mean(np.sum(X.astype(np.float64) ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_SQUARED, 0.0)
    plan = _plan(1, LOSS_SQUARED, lr, max_iter=500, target=[-1])
    W, b, _, _ = linear_fit_ref(X, y, plan)
    np.testing.assert_allclose(W[:, 0], w, atol=2e-2)
    assert abs(b[0] - 2.5) < 2e-2
    r2 = linear_scores_ref(X, y, W, b, plan, "r2")
    assert r2[0] > 0.999


def test_early_stop_tol(binary_blobs):
    X, y = binary_blobs
    msq = float(np.mean(np.sum(X ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    plan = _plan(2, LOSS_LOGISTIC, lr, max_iter=500,
                 tol=np.float32(1e-3))
    plan.tol[:] = [1e-2, 0.0]
    W, b, n_iter, _ = linear_fit_ref(X, y, plan)
    assert n_iter[0] < n_iter[1]          # looser tol stops earlier
    assert n_iter[1] == 500


def test_multiclass_ovr_scoring():
    rng = np.random.default_rng(2)
    n, f, C = 600, 10, 3
    centers = rng.standard_normal((C, f)) * 3
    y = rng.integers(0, C, n)
    X = centers[y] + rng.standard_normal((n, f))
    msq = float(np.mean(np.sum(X ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    # 2 candidates x 3 classes = 6 columns
    plan = _plan(6, LOSS_LOGISTIC, lr, max_iter=100,
                 target=[0, 1, 2, 0, 1, 2],
                 col_to_cand=[0, 0, 0, 1, 1, 1], n_classes=3)
    plan.l2[3:] = 0.5
    W, b, _, _ = linear_fit_ref(X, y, plan)
    acc = linear_scores_ref(X, y, W, b, plan, "accuracy")
    assert acc.shape == (2,)
    assert acc[0] > 0.9
    assert acc[0] >= acc[1]               # heavy reg worse on train
