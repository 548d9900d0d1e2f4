"""MI355X-native linear estimator family.

sklearn-compatible estimators (``get_params``/``set_params``/``clone``-able,
usable inside sklearn pipelines and our search classes) whose inner ``fit``
is the batched solver of ``_solver_ref`` -- executed by the hand-written
CDNA4 HIP kernels (K1/K4, ops/csrc/linear_fit.hip) when a GPU is present,
and by the numpy reference on CPU.  Replaces the sklearn Cython/BLAS inner
loops the reference borrows.

Solver note: sklearn's LogisticRegression default
solver is lbfgs; ours is deterministic full-batch GD with momentum.  Scores
converge to comparable values on standardized data (the conformance tests
compare scores, not weights); ``tol``/``max_iter`` are honored per column.
Multiclass is one-vs-rest (the sklearn 0.18/0.19-era default the reference
tracked), which is also what makes multiclass fits batchable: every
(candidate, class) pair is one column of a single batched kernel launch.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from sklearn.base import BaseEstimator, ClassifierMixin, RegressorMixin

from ._solver_ref import (LOSS_HINGE, LOSS_LOGISTIC, LOSS_SQUARED, BatchPlan,
                          auto_lr, linear_fit_ref)

__all__ = ["LogisticRegression", "LinearSVC", "SGDClassifier",
           "LinearRegression", "Ridge"]


class _LinearBase(BaseEstimator):
    """Shared plumbing for the batched linear family."""

    _hip_batched = True          # engine routes fits through ops/ kernels
    _engine_family = "linear"
    _loss_id: int = LOSS_LOGISTIC

    # -- per-estimator plan hooks (engine + local fit share these) ---------
    def _column_spec(self, n_samples: int, mean_sq_row_norm: float) -> dict:
        """Per-column solver parameters for ONE column of this estimator.

        Returns dict(loss, lr, l1, l2, momentum, tol, max_iter,
        fit_intercept).  Subclasses map their sklearn-surface params
        (C / alpha / penalty) onto (l1, l2).
        """
        raise NotImplementedError

    def _spec_common(self, l1: float, l2: float, n: int,
                     mean_sq_row_norm: float) -> dict:
        # NOTE (documented divergence): inside a search the auto lr uses
        # the FULL matrix's mean squared row norm (fold-invariant, so B
        # candidates x 5 folds share one plan); a standalone .fit() sees
        # only its own rows.  Label-free scale statistic -- no leakage of
        # fold targets -- but trajectories differ slightly between the
        # two entry points (tests/test_fuzz.py pins the invariant that
        # DOES matter: column independence within a batch).
        lr = self.learning_rate
        if lr in (None, "auto"):
            lr = auto_lr(mean_sq_row_norm, self._loss_id, l2)
        return dict(loss=self._loss_id, lr=float(lr), l1=float(l1),
                    l2=float(l2), momentum=float(self.momentum),
                    tol=float(self.tol), max_iter=int(self.max_iter),
                    fit_intercept=1 if self.fit_intercept else 0)

    # -- single-estimator fit (CPU numpy; GPU via engine helper) -----------
    def fit(self, X, y, sample_weight=None):
        from ..engine import fit_single

        return fit_single(self, X, y, sample_weight=sample_weight)

    def _finalize_fit(self, W: np.ndarray, b: np.ndarray,
                      n_iter: np.ndarray, classes: Optional[np.ndarray]):
        """Install fitted params.  W is [F, n_cols] column-major batch."""
        if classes is not None:
            self.classes_ = classes
            if len(classes) == 2:
                self.coef_ = W.T[[0]]          # [1, F], sklearn binary layout
                self.intercept_ = b[:1].astype(np.float64)
            else:
                self.coef_ = np.ascontiguousarray(W.T)   # [C, F]
                self.intercept_ = b.astype(np.float64)
        else:
            self.coef_ = W[:, 0]
            self.intercept_ = float(b[0])
        self.n_iter_ = n_iter
        self.n_features_in_ = (self.coef_.shape[-1]
                               if self.coef_.ndim else len(self.coef_))
        return self

    def decision_function(self, X):
        self._check_fitted()
        import scipy.sparse as sp

        if not sp.issparse(X):
            X = np.asarray(X, dtype=np.float64)
        coef = np.atleast_2d(self.coef_)
        scores = np.asarray(X @ coef.T) + np.atleast_1d(self.intercept_)
        return scores.ravel() if scores.shape[1] == 1 else scores

    def _check_fitted(self):
        if not hasattr(self, "coef_"):
            raise AttributeError(
                f"this {type(self).__name__} instance is not fitted yet")


class _LinearClassifier(ClassifierMixin, _LinearBase):

    def predict(self, X):
        scores = self.decision_function(X)
        if scores.ndim == 1:
            return self.classes_[(scores > 0).astype(int)]
        return self.classes_[np.argmax(scores, axis=1)]


class LogisticRegression(_LinearClassifier):
    """Logistic regression (OVR multiclass) on CDNA4 MFMA kernels.

    Surface follows sklearn: ``C`` (inverse regularization), ``penalty`` in
    {"l2", "l1", "elasticnet", "none"}, ``l1_ratio``, ``tol``, ``max_iter``,
    ``fit_intercept``.  Engine extras: ``learning_rate`` ("auto" =
    curvature-bound step), ``momentum``.
    """

    _loss_id = LOSS_LOGISTIC

    def __init__(self, C=1.0, penalty="l2", l1_ratio=0.5, tol=1e-4,
                 max_iter=100, fit_intercept=True, learning_rate="auto",
                 momentum=0.9, class_weight=None):
        self.C = C
        self.penalty = penalty
        self.l1_ratio = l1_ratio
        self.tol = tol
        self.max_iter = max_iter
        self.fit_intercept = fit_intercept
        self.learning_rate = learning_rate
        self.momentum = momentum
        self.class_weight = class_weight

    def _column_spec(self, n, mean_sq_row_norm):
        # sklearn objective: sum(logloss) + ||w||^2/(2C); per-sample scaled
        base = 1.0 / (max(float(self.C), 1e-12) * n)
        l1, l2 = _penalty_split(self.penalty, base, self.l1_ratio)
        return self._spec_common(l1, l2, n, mean_sq_row_norm)

    def predict_proba(self, X):
        scores = self.decision_function(X)
        if scores.ndim == 1:
            p = 1.0 / (1.0 + np.exp(-scores))
            return np.column_stack([1.0 - p, p])
        p = 1.0 / (1.0 + np.exp(-scores))
        return p / np.maximum(p.sum(axis=1, keepdims=True), 1e-12)

    def predict_log_proba(self, X):
        return np.log(np.maximum(self.predict_proba(X), 1e-300))


class LinearSVC(_LinearClassifier):
    """Linear SVM (hinge loss, OVR multiclass) on CDNA4 MFMA kernels."""

    _loss_id = LOSS_HINGE

    def __init__(self, C=1.0, penalty="l2", tol=1e-4, max_iter=200,
                 fit_intercept=True, learning_rate="auto", momentum=0.9,
                 class_weight=None):
        self.C = C
        self.penalty = penalty
        self.tol = tol
        self.max_iter = max_iter
        self.fit_intercept = fit_intercept
        self.learning_rate = learning_rate
        self.momentum = momentum
        self.class_weight = class_weight

    def _column_spec(self, n, mean_sq_row_norm):
        base = 1.0 / (max(float(self.C), 1e-12) * n)
        l1, l2 = _penalty_split(self.penalty, base, 0.5)
        return self._spec_common(l1, l2, n, mean_sq_row_norm)


class SGDClassifier(_LinearClassifier):
    """SGD-style linear classifier: ``loss`` in {"log_loss", "hinge",
    "squared_error"}, regularization via ``alpha``/``penalty``."""

    def __init__(self, loss="hinge", penalty="l2", alpha=1e-4, l1_ratio=0.15,
                 tol=1e-3, max_iter=100, fit_intercept=True,
                 learning_rate="auto", momentum=0.9, class_weight=None):
        self.loss = loss
        self.penalty = penalty
        self.alpha = alpha
        self.l1_ratio = l1_ratio
        self.tol = tol
        self.max_iter = max_iter
        self.fit_intercept = fit_intercept
        self.learning_rate = learning_rate
        self.momentum = momentum
        self.class_weight = class_weight

    @property
    def _loss_id(self):
        return {"log_loss": LOSS_LOGISTIC, "log": LOSS_LOGISTIC,
                "hinge": LOSS_HINGE,
                "squared_error": LOSS_SQUARED}[self.loss]

    def _column_spec(self, n, mean_sq_row_norm):
        l1, l2 = _penalty_split(self.penalty, float(self.alpha),
                                self.l1_ratio)
        return self._spec_common(l1, l2, n, mean_sq_row_norm)


class LinearRegression(RegressorMixin, _LinearBase):
    """Ordinary least squares via the batched GD solver (squared loss)."""

    _loss_id = LOSS_SQUARED

    def __init__(self, fit_intercept=True, tol=1e-6, max_iter=300,
                 learning_rate="auto", momentum=0.9):
        self.fit_intercept = fit_intercept
        self.tol = tol
        self.max_iter = max_iter
        self.learning_rate = learning_rate
        self.momentum = momentum

    def _column_spec(self, n, mean_sq_row_norm):
        return self._spec_common(0.0, 0.0, n, mean_sq_row_norm)

    def predict(self, X):
        return self.decision_function(X)


class Ridge(RegressorMixin, _LinearBase):
    """Ridge regression (squared loss + L2)."""

    _loss_id = LOSS_SQUARED

    def __init__(self, alpha=1.0, fit_intercept=True, tol=1e-6, max_iter=300,
                 learning_rate="auto", momentum=0.9):
        self.alpha = alpha
        self.fit_intercept = fit_intercept
        self.tol = tol
        self.max_iter = max_iter
        self.learning_rate = learning_rate
        self.momentum = momentum

    def _column_spec(self, n, mean_sq_row_norm):
        return self._spec_common(0.0, float(self.alpha) / n, n,
                                 mean_sq_row_norm)

    def predict(self, X):
        return self.decision_function(X)


def _penalty_split(penalty, base, l1_ratio):
    if penalty in (None, "none"):
        return 0.0, 0.0
    if penalty == "l2":
        return 0.0, base
    if penalty == "l1":
        return base, 0.0
    if penalty == "elasticnet":
        return base * l1_ratio, base * (1.0 - l1_ratio)
    raise ValueError(f"unknown penalty {penalty!r}")
