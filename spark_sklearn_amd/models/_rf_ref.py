"""Numpy reference for the device RandomForest (K3b kernels).

Replicates `ops/csrc/ska_ops.hip::rf_fit` bit-for-bit at the structure
level: the same stateless hash (rf_hash) drives the bootstrap resample and
the exact-k per-node feature subsampling, splits use the summed per-class
gini gain (== the xgboost-style gain with g=class-count, h=count, lam=0;
for binary the two-class sum collapses to twice the single-channel gain,
which leaves the argmax unchanged), and leaf values are class
probabilities / means divided by n_estimators so a summing tree walk
yields the forest average.

Documented divergences from sklearn's RandomForest*:
- splits are on 256-bin quantile-binned features (same binning as the GBT
  family), not exact midpoints;
- max_features selects the k features with the smallest per-(tree, node)
  hash values -- the same uniform k-subset distribution as sklearn's
  permutation draw, but a different (deterministic) stream;
- tree growth is depth-wise with max_depth required (device cap 10).
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np

__all__ = ["RFSpec", "rf_fit_ref", "rf_hash", "rf_predict_ref"]

_U32 = np.uint32


@dataclass
class RFSpec:
    n_estimators: int = 100
    max_depth: int = 10
    min_samples_leaf: int = 1
    max_features: int = 0          # 0 => all features
    bootstrap: bool = True
    random_state: int = 0
    n_classes: int = 0             # 0/2 => single channel; >2 => planes
    splitter: str = "best"         # "random" => ExtraTrees thresholds


def rf_hash(s):
    """splitmix32-style avalanche; matches the HIP rf_hash exactly
    (uint32 wrap-around multiplication is intentional)."""
    with np.errstate(over="ignore"):
        s = _U32(s) if np.isscalar(s) else s.astype(_U32)
        s = s ^ (s >> _U32(16))
        s = (s * _U32(0x7FEB352D)).astype(_U32)
        s = s ^ (s >> _U32(15))
        s = (s * _U32(0x846CA68B)).astype(_U32)
        s = s ^ (s >> _U32(16))
        return s


def _feat_hash(fseed, nd, f):
    with np.errstate(over="ignore"):
        return rf_hash(
            _U32(fseed) ^ (_U32(nd) * _U32(2654435761)) ^
            (f.astype(_U32) * _U32(40503)))


def _bootstrap_positions(nt, seed):
    i = np.arange(nt, dtype=np.uint64)
    r = rf_hash((_U32(seed) ^ i.astype(_U32)))
    return ((r.astype(np.uint64) * np.uint64(nt)) >> np.uint64(32)).astype(
        np.int64)


def _thr_hash(fseed, nd, f):
    with np.errstate(over="ignore"):
        return rf_hash(_feat_hash(fseed, nd, np.asarray([f]))[0] ^
                       _U32(0x9E3779B9))


def _selected_features(F, kfeat, fseed, nd):
    if kfeat <= 0 or kfeat >= F:
        return np.ones(F, dtype=bool)
    h = _feat_hash(fseed, nd, np.arange(F))
    sel = np.zeros(F, dtype=bool)
    sel[np.argsort(h, kind="stable")[:kfeat]] = True
    return sel


def _build_rf_tree(Xb, gplanes, spec, fseed, w=None):
    """One tree; gplanes [KP][n] class-indicator (or y) channels.

    w (per-row sample_weight, already applied to gplanes by the caller):
    gini/MSE denominators and leaf values use the weighted mass sum(w),
    while min_samples_leaf keeps plain row COUNTS (sklearn semantics --
    weighted leaf mass is min_weight_fraction_leaf territory)."""
    n, F = Xb.shape
    KP = gplanes.shape[0]
    NN = 2 ** (spec.max_depth + 1) - 1
    feat = np.full(NN, -1, dtype=np.int32)
    thr = np.zeros(NN, dtype=np.uint8)
    val = np.zeros((KP, NN), dtype=np.float32)
    node = np.zeros(n, dtype=np.int32)
    alive = np.ones(n, dtype=bool)
    eps = 1e-12
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
            cnt = len(rows)
            G = gplanes[:, rows].sum(axis=1)            # [KP]
            if KP > 1:
                # classes partition the mass: the denominator is the SUM
                # of the class numerators (same accumulation), so pure
                # nodes cancel to exactly zero gain -- mirrors the
                # device split kernel, which derives the mass from the
                # class-g prefixes for the same reason
                wmass = float(np.sum(G))
            elif w is None:
                wmass = cnt
            else:
                wmass = float(np.sum(w[rows]))
            parent = float(np.sum(G * G) / (wmass + eps))
            sel = _selected_features(F, spec.max_features, fseed, nd)
            best_gain, best_f, best_t = 0.0, -1, 0
            for f in range(F):
                if not sel[f]:
                    continue
                bins = Xb[rows, f]
                ch = np.bincount(bins, minlength=256)
                cl = np.cumsum(ch)[:-1].astype(np.float64)
                cr = cnt - cl
                gls = []
                for c in range(KP):
                    gh = np.bincount(bins, weights=gplanes[c, rows],
                                     minlength=256)
                    gls.append(np.cumsum(gh)[:-1])
                if KP > 1:
                    hl = np.sum(gls, axis=0)
                    hr = wmass - hl
                elif w is None:
                    hl, hr = cl, cr
                else:
                    hh = np.bincount(bins, weights=w[rows], minlength=256)
                    hl = np.cumsum(hh)[:-1]
                    hr = wmass - hl
                # factored form (matches the device kernel): the gini
                # denominators are class-independent, so divide once
                A = np.zeros(255)
                Bq = np.zeros(255)
                for c in range(KP):
                    gl = gls[c]
                    gr = G[c] - gl
                    A += gl * gl
                    Bq += gr * gr
                gain = A / (hl + eps) + Bq / (hr + eps) - parent
                okay = (cl >= spec.min_samples_leaf) & \
                    (cr >= spec.min_samples_leaf)
                if spec.splitter == "random":
                    # ExtraTrees: one uniform threshold bin over the
                    # node's occupied [minbin, maxbin) range
                    nz = np.flatnonzero(ch)
                    if len(nz) < 2:
                        continue
                    lo, hi = int(nz[0]), int(nz[-1])
                    tstar = lo + int(_thr_hash(fseed, nd, f) %
                                     _U32(hi - lo))
                    mask = np.zeros(255, dtype=bool)
                    mask[tstar] = True
                    okay &= mask
                gain = np.where(okay, gain, -np.inf)
                t = int(np.argmax(gain))
                if gain[t] > best_gain + 1e-12:
                    best_gain, best_f, best_t = float(gain[t]), f, t
            if best_f < 0:
                feat[nd] = -1
                val[:, nd] = (G / (wmass + eps)).astype(np.float32)
                alive[rows] = False
                continue
            feat[nd] = best_f
            thr[nd] = best_t
            node[rows] = 2 * nd + 1 + (Xb[rows, best_f] > best_t
                                       ).astype(np.int32)
    live = np.flatnonzero(alive)
    for nd in np.unique(node[live]):
        rows = live[node[live] == nd]
        G = gplanes[:, rows].sum(axis=1)
        if KP > 1:
            wmass = float(np.sum(G))
        elif w is None:
            wmass = len(rows)
        else:
            wmass = float(np.sum(w[rows]))
        feat[nd] = -1
        val[:, nd] = (G / (wmass + eps)).astype(np.float32)
    return feat, thr, val


def rf_fit_ref(Xb: np.ndarray, y: np.ndarray, spec: RFSpec,
               sample_weight=None):
    """Returns trees in the gbt walk format.

    Binary/regression: list of (feat, thr, val); K>2: list of per-class
    (feat, thr, val_c) lists (planes share feat/thr).  Leaf values are
    divided by n_estimators.
    """
    n = Xb.shape[0]
    KP = spec.n_classes if spec.n_classes > 2 else 1
    trees = []
    for t in range(spec.n_estimators):
        with np.errstate(over="ignore"):
            st = int(rf_hash(_U32(spec.random_state) ^
                             (_U32(t) * _U32(0x9E3779B9))))
        fs = int(rf_hash(_U32(st) ^ _U32(0x85EBCA6B)))
        if spec.bootstrap:
            pos = _bootstrap_positions(n, st)
        else:
            pos = np.arange(n)
        Xb_t = Xb[pos]
        y_t = y[pos]
        w_t = None if sample_weight is None else \
            np.asarray(sample_weight, np.float64)[pos]
        if KP > 1:
            gpl = np.stack([(y_t == c).astype(np.float32)
                            for c in range(KP)])
        else:
            gpl = y_t.astype(np.float32)[None, :]
        if w_t is not None:
            gpl = (gpl * w_t[np.newaxis, :]).astype(np.float32)
        feat, thr, val = _build_rf_tree(Xb_t, gpl, spec, fs, w_t)
        val = val / np.float32(spec.n_estimators)
        if KP > 1:
            trees.append([(feat, thr, val[c]) for c in range(KP)])
        else:
            trees.append((feat, thr, val[0]))
    return trees


def rf_predict_ref(Xb: np.ndarray, trees, n_classes: int = 0):
    """Forest-average prediction: probability of class 1 (binary), the
    per-class probability matrix (K>2) or the mean response."""
    from ._gbt_ref import _tree_apply

    if n_classes > 2:
        out = np.zeros((Xb.shape[0], n_classes), dtype=np.float32)
        for planes in trees:
            for c, (f, t, v) in enumerate(planes):
                out[:, c] += _tree_apply(Xb, f, t, v)
        return out
    out = np.zeros(Xb.shape[0], dtype=np.float32)
    for f, t, v in trees:
        out += _tree_apply(Xb, f, t, v)
    return out
