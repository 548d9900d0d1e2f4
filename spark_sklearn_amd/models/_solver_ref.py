"""Reference batched linear-model solver (numpy, fp32).

This module defines the EXACT numerical semantics of the K1 HIP kernel
family (``ops/csrc/linear_fit.hip``): the device kernels are tested against
this implementation (bf16-compute / fp32-accum tolerances) and the CPU tier
of the search runs it directly.  SURVEY.md section 2.5 (K1) / section 7.3
("sklearn numeric parity": compare scores, not weights).

Design: ONE solver for the whole linear family (LogisticRegression,
LinearSVC, SGDClassifier, Ridge/LinearRegression GD path), batched over B
*columns*.  A column is one (candidate, OVR-class) pair: a candidate on a
C-class problem contributes C columns (C=1 for binary / regression), so
heterogeneous grids -- even mixed estimator families with different losses
per candidate (BASELINE config 3) -- fold into a single batch with
per-column loss/lr/regularization, which is what lets a lone 256-feature
fit feed 256 CUs.

Optimizer: full-batch gradient descent with classical momentum and
per-column early stop on relative loss change.  Deterministic by
construction (no minibatch shuffling), which makes cross-world-size
determinism tests meaningful.

Losses (t = per-column target derived from labels on the fly):
  logistic: t in {0,1};  d = sigmoid(z) - t;     loss = logloss
  hinge:    s in {-1,1}; d = -s if s*z < 1 else 0; loss = max(0, 1-s*z)
  squared:  d = z - y;                            loss = 0.5*(z-y)^2

Gradient: gW = X^T d / n + l2*W + l1*sign(W);  gb = mean(d)
Update:   V = momentum*V - lr*g;  W += V
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np

__all__ = ["BatchPlan", "linear_fit_ref", "linear_scores_ref",
           "LOSS_LOGISTIC", "LOSS_HINGE", "LOSS_SQUARED", "auto_lr"]

LOSS_LOGISTIC = 0
LOSS_HINGE = 1
LOSS_SQUARED = 2


@dataclass
class BatchPlan:
    """Per-column kernel parameters for one candidate batch.

    All arrays have length B (number of columns).  ``target_class`` is the
    encoded class index a column discriminates (-1 for regression columns);
    ``col_to_cand`` maps columns to candidate slots; ``n_classes`` is the
    number of OVR columns per candidate (1 = binary/regression).
    """

    loss: np.ndarray           # int32 [B]
    lr: np.ndarray             # float32 [B]
    l1: np.ndarray             # float32 [B]
    l2: np.ndarray             # float32 [B]
    momentum: np.ndarray       # float32 [B]
    tol: np.ndarray            # float32 [B]
    max_iter: np.ndarray       # int32 [B]
    target_class: np.ndarray   # int32 [B]
    col_to_cand: np.ndarray    # int32 [B]
    n_classes: int = 1
    fit_intercept: np.ndarray = None  # uint8 [B]

    @property
    def B(self) -> int:
        return len(self.loss)


def auto_lr(mean_sq_row_norm: float, loss_id: int, l2: float) -> float:
    """Step size from a curvature bound: L ~= c_loss * E[||x||^2] + l2."""
    curvature = 0.25 if loss_id == LOSS_LOGISTIC else 1.0
    return 1.0 / (curvature * mean_sq_row_norm + l2 + 1e-12)


def _column_targets(y: np.ndarray, plan: BatchPlan) -> np.ndarray:
    """Materialize per-column targets [n, B] from labels + plan (the HIP
    kernel computes these on the fly from y + target_class)."""
    n = y.shape[0]
    T = np.empty((n, plan.B), dtype=np.float32)
    for j in range(plan.B):
        if plan.loss[j] == LOSS_SQUARED and plan.target_class[j] < 0:
            T[:, j] = y.astype(np.float32)
        elif plan.loss[j] == LOSS_HINGE or plan.loss[j] == LOSS_SQUARED:
            T[:, j] = np.where(y == plan.target_class[j], 1.0, -1.0)
        else:  # logistic: {0,1}
            T[:, j] = (y == plan.target_class[j]).astype(np.float32)
    return T


def linear_fit_ref(X: np.ndarray, y: np.ndarray, plan: BatchPlan,
                   sample_stride: int = 1,
                   sample_weight: Optional[np.ndarray] = None):
    """Fit B columns simultaneously.  Returns (W [F,B], b [B], n_iter [B],
    final_loss [B]).  X may be dense or scipy CSR.

    sample_weight (already normalized to mean ~1 by the engine) scales
    each row's loss AND gradient: d_i -> w_i * d_i, l_i -> w_i * l_i.
    Everything downstream (G = X^T D / n, gb = mean(D), the convergence
    loss) then weights automatically.  The device kernels apply the same
    per-row multiply in the fwd epilogue (fwd_d3/fwd_d2 WT=1)."""
    import scipy.sparse as sp

    if sp.issparse(X):
        X = sp.csr_matrix(X, dtype=np.float32)
    else:
        X = np.ascontiguousarray(X, dtype=np.float32)
    n, F = X.shape
    B = plan.B
    T = _column_targets(np.asarray(y), plan)
    w_row = None
    if sample_weight is not None:
        w_row = np.asarray(sample_weight, dtype=np.float32).reshape(n, 1)

    W = np.zeros((F, B), dtype=np.float32)
    b = np.zeros(B, dtype=np.float32)
    V = np.zeros((F, B), dtype=np.float32)
    Vb = np.zeros(B, dtype=np.float32)
    active = np.ones(B, dtype=bool)
    prev_loss = np.full(B, np.inf, dtype=np.float64)
    n_iter = np.zeros(B, dtype=np.int32)
    cur_loss = np.zeros(B, dtype=np.float64)

    max_epochs = int(plan.max_iter.max())
    inv_n = np.float32(1.0 / n)
    for epoch in range(max_epochs):
        Z = np.asarray(X @ W) + b[np.newaxis, :]        # [n, B]
        D = np.empty_like(Z)
        loss_sum = np.zeros(B, dtype=np.float64)
        for lid, cols in _loss_groups(plan):
            z, t = Z[:, cols], T[:, cols]
            if lid == LOSS_LOGISTIC:
                p = 1.0 / (1.0 + np.exp(-z))
                D[:, cols] = p - t
                eps = 1e-12
                le = -(t * np.log(p + eps) + (1 - t) * np.log(1 - p + eps))
            elif lid == LOSS_HINGE:
                margin = t * z
                D[:, cols] = np.where(margin < 1.0, -t, 0.0)
                le = np.maximum(0.0, 1.0 - margin)
            else:
                r = z - t
                D[:, cols] = r
                le = 0.5 * r * r
            if w_row is not None:
                le = le * w_row
            loss_sum[cols] = np.sum(le, axis=0)
        if w_row is not None:
            D *= w_row

        # data-loss only (no reg term): the convergence criterion the HIP
        # finalize kernel uses -- keeps device/host n_iter_ identical
        # without a cross-column ||W|| reduction on device
        cur = loss_sum * inv_n
        epoch_active = active & (epoch < plan.max_iter)
        cur_loss = np.where(epoch_active, cur, cur_loss)
        n_iter = np.where(epoch_active, epoch + 1, n_iter).astype(np.int32)

        G = np.asarray(X.T @ D) * inv_n + plan.l2 * W \
            + plan.l1 * np.sign(W)                           # [F, B]
        gb = D.mean(axis=0) * plan.fit_intercept

        mask = epoch_active.astype(np.float32)
        # masked momentum update: inactive columns freeze V and W entirely
        # (identical to the device kernels, which skip frozen columns)
        V = np.where(mask, plan.momentum * V - plan.lr * G, V)
        Vb = np.where(mask, plan.momentum * Vb - plan.lr * gb, Vb)
        W += V * mask
        b += Vb * mask

        converged = np.abs(prev_loss - cur) <= plan.tol * np.maximum(cur, 1.0)
        active = active & ~(converged & epoch_active)
        prev_loss = np.where(epoch_active, cur, prev_loss)
        if not active.any():
            break
    return W, b, n_iter, cur_loss


def _loss_groups(plan: BatchPlan):
    for lid in (LOSS_LOGISTIC, LOSS_HINGE, LOSS_SQUARED):
        cols = np.flatnonzero(plan.loss == lid)
        if len(cols):
            yield lid, cols


def linear_scores_ref(X: np.ndarray, y: np.ndarray, W: np.ndarray,
                      b: np.ndarray, plan: BatchPlan,
                      metric: str = "accuracy",
                      classes: Optional[np.ndarray] = None) -> np.ndarray:
    """Per-candidate scores [n_cand] for the fitted batch (K4 semantics).

    metric: 'accuracy' (OVR argmax / sign), 'neg_log_loss', or 'r2'.
    """
    import scipy.sparse as sp

    if not sp.issparse(X):
        X = np.ascontiguousarray(X, dtype=np.float32)
    Z = np.asarray(X @ W) + b[np.newaxis, :]
    n_cand = int(plan.col_to_cand.max()) + 1
    C = plan.n_classes
    scores = np.zeros(n_cand, dtype=np.float64)
    y = np.asarray(y)

    if metric == "r2":
        ss_tot = float(np.sum((y - y.mean()) ** 2))
        for cand in range(n_cand):
            col = np.flatnonzero(plan.col_to_cand == cand)[0]
            ss_res = float(np.sum((y - Z[:, col]) ** 2))
            scores[cand] = 1.0 - ss_res / ss_tot if ss_tot > 0 else 0.0
        return scores

    if C == 1:  # binary: column discriminates classes[1] vs classes[0]
        for cand in range(n_cand):
            col = np.flatnonzero(plan.col_to_cand == cand)[0]
            if metric == "accuracy":
                if plan.loss[col] == LOSS_LOGISTIC:
                    pred_pos = Z[:, col] > 0.0
                else:
                    pred_pos = Z[:, col] > 0.0
                pred = np.where(pred_pos, plan.target_class[col],
                                _other_class(plan, col, classes))
                scores[cand] = float(np.mean(pred == y))
            else:  # neg_log_loss, binary logistic
                p = 1.0 / (1.0 + np.exp(-Z[:, col]))
                t = (y == plan.target_class[col]).astype(np.float64)
                eps = 1e-12
                scores[cand] = float(np.mean(
                    t * np.log(p + eps) + (1 - t) * np.log(1 - p + eps)))
        return scores

    # multiclass OVR: candidate's C contiguous columns; argmax decides
    for cand in range(n_cand):
        cols = np.flatnonzero(plan.col_to_cand == cand)
        z = Z[:, cols]
        if metric == "accuracy":
            pred_idx = np.argmax(z, axis=1)
            pred = plan.target_class[cols][pred_idx]
            scores[cand] = float(np.mean(pred == y))
        else:
            p = 1.0 / (1.0 + np.exp(-z))
            p = p / np.maximum(p.sum(axis=1, keepdims=True), 1e-12)
            t_idx = np.searchsorted(plan.target_class[cols], y)
            eps = 1e-12
            scores[cand] = float(np.mean(
                np.log(p[np.arange(len(y)), t_idx] + eps)))
    return scores


def _other_class(plan: BatchPlan, col: int, classes):
    if classes is None or len(classes) != 2:
        return 1 - plan.target_class[col]
    cls = np.asarray(classes)
    return cls[0] if plan.target_class[col] == cls[1] else cls[1]
