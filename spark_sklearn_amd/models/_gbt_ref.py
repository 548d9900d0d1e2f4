"""Reference histogram gradient-boosting (numpy) -- the K3 kernel spec.

Defines the exact semantics of the GBT HIP kernels (SURVEY.md section 2.5
K3): LightGBM/XGBoost-style HISTOGRAM boosting -- features quantile-binned
to <=256 bins once per dataset (uint8 matrix), trees grown depth-wise in
heap layout, per-level per-(node, feature) gradient/hessian histograms,
best-split scan with the XGBoost gain formula, leaf values -G/(H+lambda).

Losses: logistic (binary classification; g = p - y, h = p(1-p)) and
squared (regression; g = pred - y, h = 1).

Documented divergences from sklearn's exact-split GradientBoosting*
(SURVEY.md section 7.3 policy -- scores, not internals): histogram splits
(matching sklearn's own HistGradientBoosting*), bins computed on the full
matrix once per dataset rather than per CV fold (binning leakage is
negligible and re-binning per fold would dominate runtime).
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np

__all__ = ["GBTSpec", "quantile_bins", "bin_data", "gbt_fit_ref",
           "gbt_fit_ref_multi", "gbt_predict_raw", "gbt_predict_raw_multi",
           "sigmoid", "softmax_rows"]

LOSS_LOGISTIC = 0
LOSS_SQUARED = 2


@dataclass
class GBTSpec:
    loss: int = LOSS_LOGISTIC
    n_estimators: int = 100
    learning_rate: float = 0.1
    max_depth: int = 3
    min_samples_leaf: int = 20
    l2_regularization: float = 0.0
    max_bins: int = 256


def sigmoid(z):
    return 1.0 / (1.0 + np.exp(-z))


def quantile_bins(X: np.ndarray, max_bins: int = 256,
                  sample_rows: int = 200_000, seed: int = 0):
    """Per-feature bin edges (<= max_bins-1 thresholds), from a sample."""
    X = np.asarray(X)
    n = X.shape[0]
    if n > sample_rows:
        rng = np.random.default_rng(seed)
        X = X[rng.choice(n, sample_rows, replace=False)]
    edges = []
    qs = np.linspace(0, 1, max_bins + 1)[1:-1]
    for f in range(X.shape[1]):
        e = np.unique(np.quantile(X[:, f], qs))
        edges.append(e.astype(np.float64))
    return edges


def bin_data(X: np.ndarray, edges) -> np.ndarray:
    """Map rows to uint8 bin indices: bin = searchsorted(edges, x)."""
    X = np.asarray(X)
    out = np.empty(X.shape, dtype=np.uint8)
    for f in range(X.shape[1]):
        out[:, f] = np.searchsorted(edges[f], X[:, f], side="left")
    return out


def _grad_hess(loss, raw, y, w=None):
    if loss == LOSS_LOGISTIC:
        p = sigmoid(raw)
        g, h = p - y, p * (1.0 - p)
    else:
        g, h = raw - y, np.ones_like(raw)
    if w is not None:
        g, h = g * w, h * w
    return g.astype(np.float32), h.astype(np.float32)


def _init_raw(loss, y, w=None):
    m = float(np.mean(y)) if w is None else \
        float(np.sum(w * y) / max(np.sum(w), 1e-12))
    if loss == LOSS_LOGISTIC:
        p = np.clip(m, 1e-6, 1 - 1e-6)
        return float(np.log(p / (1.0 - p)))
    return m


def gbt_fit_ref(Xb: np.ndarray, y: np.ndarray, spec: GBTSpec,
                sample_weight=None):
    """Fit one model on binned data.  Returns (trees, f0).

    Tree heap layout per tree (size 2^(max_depth+1) - 1):
      feature int32 (-1 => leaf), threshold_bin uint8 (go left if
      bin <= thr), value float32 (leaf contribution, already scaled by
      the learning rate).

    sample_weight scales g AND h per row (xgboost semantics: leaf value
    -G/H and split gains become weighted automatically) plus the f0
    base score; min_samples_leaf stays a plain row COUNT (sklearn
    semantics).
    """
    n, F = Xb.shape
    y = np.asarray(y, dtype=np.float32)
    w = None if sample_weight is None else \
        np.asarray(sample_weight, dtype=np.float32)
    f0 = _init_raw(spec.loss, y, w)
    raw = np.full(n, f0, dtype=np.float32)
    n_nodes = 2 ** (spec.max_depth + 1) - 1
    lam = float(spec.l2_regularization)
    trees = []

    for _ in range(spec.n_estimators):
        g, h = _grad_hess(spec.loss, raw, y, w)
        feat = np.full(n_nodes, -1, dtype=np.int32)
        thr = np.zeros(n_nodes, dtype=np.uint8)
        val = np.zeros(n_nodes, dtype=np.float32)
        node = np.zeros(n, dtype=np.int32)          # heap position per row
        alive = np.ones(n, dtype=bool)              # still descending

        for depth in range(spec.max_depth):
            lvl_lo = 2 ** depth - 1
            lvl_hi = 2 ** (depth + 1) - 1
            live = np.flatnonzero(alive)
            if len(live) == 0:
                break
            nodes_here = np.unique(node[live])
            nodes_here = nodes_here[(nodes_here >= lvl_lo) &
                                    (nodes_here < lvl_hi)]
            for nd in nodes_here:
                rows = live[node[live] == nd]
                G, H, cnt = float(g[rows].sum()), float(h[rows].sum()), \
                    len(rows)
                parent_score = G * G / (H + lam + 1e-12)
                best_gain, best_f, best_t = 0.0, -1, 0
                best_stats = None
                for f in range(F):
                    bins = Xb[rows, f]
                    gh = np.bincount(bins, weights=g[rows], minlength=256)
                    hh = np.bincount(bins, weights=h[rows], minlength=256)
                    ch = np.bincount(bins, minlength=256)
                    gl = np.cumsum(gh)[:-1]
                    hl = np.cumsum(hh)[:-1]
                    cl = np.cumsum(ch)[:-1]
                    gr, hr, cr = G - gl, H - hl, cnt - cl
                    okay = (cl >= spec.min_samples_leaf) & \
                        (cr >= spec.min_samples_leaf)
                    gain = np.where(
                        okay,
                        gl * gl / (hl + lam + 1e-12) +
                        gr * gr / (hr + lam + 1e-12) - parent_score,
                        -np.inf)
                    t = int(np.argmax(gain))
                    if gain[t] > best_gain + 1e-12:
                        best_gain, best_f, best_t = float(gain[t]), f, t
                        best_stats = (gl[t], hl[t])
                if best_f < 0:
                    # leaf: settle value, rows stop descending
                    feat[nd] = -1
                    val[nd] = -G / (H + lam + 1e-12) * spec.learning_rate
                    alive[rows] = False
                    continue
                feat[nd] = best_f
                thr[nd] = best_t
                go_right = Xb[rows, best_f] > best_t
                node[rows] = 2 * nd + 1 + go_right.astype(np.int32)

        # bottom-level nodes become leaves
        live = np.flatnonzero(alive)
        for nd in np.unique(node[live]):
            rows = live[node[live] == nd]
            G, H = float(g[rows].sum()), float(h[rows].sum())
            feat[nd] = -1
            val[nd] = -G / (H + lam + 1e-12) * spec.learning_rate
        # apply tree to raw predictions
        raw += _tree_apply(Xb, feat, thr, val)
        trees.append((feat, thr, val))
    return trees, f0


def _tree_apply(Xb, feat, thr, val):
    n = Xb.shape[0]
    node = np.zeros(n, dtype=np.int32)
    done = np.zeros(n, dtype=bool)
    out = np.zeros(n, dtype=np.float32)
    max_nodes = len(feat)
    while True:
        leaf_mask = ~done & (feat[node] < 0)
        out[leaf_mask] = val[node[leaf_mask]]
        done |= leaf_mask
        active = ~done
        if not active.any():
            break
        nd = node[active]
        f = feat[nd]
        go_right = Xb[np.flatnonzero(active), f] > thr[nd]
        child = 2 * nd + 1 + go_right.astype(np.int32)
        if (child >= max_nodes).any():   # safety: treat as leaf
            bad = child >= max_nodes
            rows = np.flatnonzero(active)[bad]
            out[rows] = 0.0
            done[rows] = True
            child = np.minimum(child, max_nodes - 1)
        node[active] = child
    return out


def gbt_predict_raw(Xb: np.ndarray, trees, f0: float) -> np.ndarray:
    raw = np.full(Xb.shape[0], f0, dtype=np.float32)
    for feat, thr, val in trees:
        raw += _tree_apply(Xb, feat, thr, val)
    return raw


# ---------------------------------------------------------------------------
# multiclass (K > 2): softmax boosting -- one tree per class per iteration
# (sklearn GradientBoostingClassifier's multinomial deviance scheme)
# ---------------------------------------------------------------------------

def softmax_rows(raw: np.ndarray) -> np.ndarray:
    z = raw - raw.max(axis=1, keepdims=True)
    e = np.exp(z)
    return e / e.sum(axis=1, keepdims=True)


def _fit_one_tree(Xb, g, h, spec):
    """One depth-wise tree on the given gradients; returns (feat, thr,
    val, contribution)."""
    one = GBTSpec(loss=spec.loss, n_estimators=1,
                  learning_rate=spec.learning_rate,
                  max_depth=spec.max_depth,
                  min_samples_leaf=spec.min_samples_leaf,
                  l2_regularization=spec.l2_regularization,
                  max_bins=spec.max_bins)
    # reuse the single-tree builder by faking the gradient closure
    return _build_tree(Xb, g, h, one)


def _build_tree(Xb, g, h, spec):
    n, F = Xb.shape
    n_nodes = 2 ** (spec.max_depth + 1) - 1
    lam = float(spec.l2_regularization)
    feat = np.full(n_nodes, -1, dtype=np.int32)
    thr = np.zeros(n_nodes, dtype=np.uint8)
    val = np.zeros(n_nodes, dtype=np.float32)
    node = np.zeros(n, dtype=np.int32)
    alive = np.ones(n, dtype=bool)
    for depth in range(spec.max_depth):
        lvl_lo, lvl_hi = 2 ** depth - 1, 2 ** (depth + 1) - 1
        live = np.flatnonzero(alive)
        if len(live) == 0:
            break
        nodes_here = np.unique(node[live])
        nodes_here = nodes_here[(nodes_here >= lvl_lo) &
                                (nodes_here < lvl_hi)]
        for nd in nodes_here:
            rows = live[node[live] == nd]
            G, H, cnt = float(g[rows].sum()), float(h[rows].sum()), len(rows)
            parent_score = G * G / (H + lam + 1e-12)
            best_gain, best_f, best_t = 0.0, -1, 0
            for f in range(F):
                bins = Xb[rows, f]
                gh = np.bincount(bins, weights=g[rows], minlength=256)
                hh = np.bincount(bins, weights=h[rows], minlength=256)
                ch = np.bincount(bins, minlength=256)
                gl, hl, cl = np.cumsum(gh)[:-1], np.cumsum(hh)[:-1],                     np.cumsum(ch)[:-1]
                gr, hr, cr = G - gl, H - hl, cnt - cl
                okay = (cl >= spec.min_samples_leaf) &                     (cr >= spec.min_samples_leaf)
                gain = np.where(
                    okay, gl * gl / (hl + lam + 1e-12) +
                    gr * gr / (hr + lam + 1e-12) - parent_score, -np.inf)
                t = int(np.argmax(gain))
                if gain[t] > best_gain + 1e-12:
                    best_gain, best_f, best_t = float(gain[t]), f, t
            if best_f < 0:
                feat[nd] = -1
                val[nd] = -G / (H + lam + 1e-12) * spec.learning_rate
                alive[rows] = False
                continue
            feat[nd] = best_f
            thr[nd] = best_t
            node[rows] = 2 * nd + 1 + (Xb[rows, best_f] > best_t
                                       ).astype(np.int32)
    live = np.flatnonzero(alive)
    for nd in np.unique(node[live]):
        rows = live[node[live] == nd]
        G, H = float(g[rows].sum()), float(h[rows].sum())
        feat[nd] = -1
        val[nd] = -G / (H + lam + 1e-12) * spec.learning_rate
    return feat, thr, val, _tree_apply(Xb, feat, thr, val)


def gbt_fit_ref_multi(Xb: np.ndarray, y_enc: np.ndarray, K: int,
                      spec: GBTSpec, sample_weight=None):
    """Softmax boosting: trees[t][c].  Returns (trees, f0 [K])."""
    n = Xb.shape[0]
    y_enc = np.asarray(y_enc, dtype=np.int64)
    w = None if sample_weight is None else \
        np.asarray(sample_weight, dtype=np.float32)
    if w is None:
        prior = np.bincount(y_enc, minlength=K) / n
    else:
        prior = np.bincount(y_enc, weights=w, minlength=K) / \
            max(float(w.sum()), 1e-12)
    f0 = np.log(np.clip(prior, 1e-6, None)).astype(np.float32)
    raw = np.tile(f0, (n, 1))
    onehot = np.eye(K, dtype=np.float32)[y_enc]
    trees = []
    for _ in range(spec.n_estimators):
        P = softmax_rows(raw)
        round_trees = []
        for c in range(K):
            g = (P[:, c] - onehot[:, c]).astype(np.float32)
            h = (P[:, c] * (1.0 - P[:, c])).astype(np.float32)
            if w is not None:
                g, h = g * w, h * w
            feat, thr, val, contrib = _build_tree(Xb, g, h, spec)
            raw[:, c] += contrib
            round_trees.append((feat, thr, val))
        trees.append(round_trees)
    return trees, f0


def gbt_predict_raw_multi(Xb: np.ndarray, trees, f0: np.ndarray):
    n = Xb.shape[0]
    K = len(f0)
    raw = np.tile(np.asarray(f0, dtype=np.float32), (n, 1))
    for round_trees in trees:
        for c, (feat, thr, val) in enumerate(round_trees):
            raw[:, c] += _tree_apply(Xb, feat, thr, val)
    return raw
