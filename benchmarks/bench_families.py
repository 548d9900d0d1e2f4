#!/usr/bin/env python3
"""Secondary family benchmarks (BASELINE configs 3-5 evidence).

Prints one JSON line per family.  Run on an MI355X box:
    python benchmarks/bench_families.py
"""

import json
import os
import sys
import time

sys.path.insert(0, "/root/repo")

import numpy as np


def bench_kmeans():
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import KMeans

    rng = np.random.default_rng(0)
    n, f, true_k = int(os.environ.get("KM_ROWS", "1000000")), 256, 16
    centers = rng.standard_normal((true_k, f)).astype(np.float32) * 3
    lab = rng.integers(0, true_k, n)
    X = centers[lab] + rng.standard_normal((n, f), dtype=np.float32)

    grid = {"n_clusters": [4, 8, 12, 16, 24, 32], "n_init": [2]}
    gs = ska.GridSearchCV(None, KMeans(random_state=0, max_iter=20, tol=0.0),
                          grid, cv=5, return_train_score=False)
    t0 = time.perf_counter()
    gs.fit(X)
    dt = time.perf_counter() - t0
    return {"family": "kmeans",
            "config": f"{n}x256, k in 4..32 x n_init=2, cv=5, 20 Lloyd "
                      "iters", "n_fits": gs.n_fits_,
            "wall_s": round(dt, 2),
            "fits_per_sec": round(gs.n_fits_ / dt, 2),
            "best": gs.best_params_}


def bench_gbt():
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import GradientBoostingClassifier

    rng = np.random.default_rng(0)
    n, f = int(os.environ.get("GBT_ROWS", "1000000")), 128
    X = rng.standard_normal((n, f), dtype=np.float32)
    y = ((X[:, 0] * X[:, 1] + X[:, 2] + X[:, 3] ** 2 - 1 +
          0.5 * rng.standard_normal(n)) > 0).astype(np.int64)

    grid = {"max_depth": [3, 5], "learning_rate": [0.1, 0.3]}
    gs = ska.GridSearchCV(None,
                          GradientBoostingClassifier(n_estimators=50),
                          grid, cv=3, return_train_score=False)
    t0 = time.perf_counter()
    gs.fit(X, y)
    dt = time.perf_counter() - t0
    return {"family": "gbt", "config": f"{n}x128, 50 trees, depth 3/5, cv=3",
            "n_fits": gs.n_fits_, "wall_s": round(dt, 2),
            "fits_per_sec": round(gs.n_fits_ / dt, 3),
            "trees_per_sec": round(gs.n_fits_ * 50 / dt, 1),
            "best": gs.best_params_, "best_acc": round(gs.best_score_, 4)}


def bench_gbt_multi():
    """Multiclass GBT: one tree per class per round (softmax boosting)."""
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import GradientBoostingClassifier

    rng = np.random.default_rng(0)
    n, f, K = 1_000_000, 64, 8
    centers = rng.standard_normal((K, f)).astype(np.float32) * 1.5
    y = rng.integers(0, K, n)
    X = centers[y] + rng.standard_normal((n, f), dtype=np.float32)

    gs = ska.GridSearchCV(None,
                          GradientBoostingClassifier(n_estimators=20),
                          {"max_depth": [3, 5]}, cv=3,
                          return_train_score=False)
    t0 = time.perf_counter()
    gs.fit(X, y)
    dt = time.perf_counter() - t0
    trees = gs.n_fits_ * 20 * K
    return {"family": "gbt-multiclass",
            "config": f"1Mx{f}, K={K}, 20 rounds, depth 3/5, cv=3",
            "n_fits": gs.n_fits_, "wall_s": round(dt, 2),
            "trees_per_sec": round(trees / dt, 1),
            "best_acc": round(gs.best_score_, 4)}


def bench_sweep():
    """Config 5: large randomized sweep with successive-halving pruning."""
    from scipy.stats import loguniform

    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    n, f = int(os.environ.get("KM_ROWS", "1000000")), 256
    X = rng.standard_normal((n, f), dtype=np.float32)
    w = rng.standard_normal(f, dtype=np.float32) / np.sqrt(f)
    y = ((X @ w + 0.25 * rng.standard_normal(n, dtype=np.float32)) > 0
         ).astype(np.int64)

    n_iter = int(os.environ.get("SWEEP_N", "10000"))
    rs = ska.RandomizedSearchCV(
        None, LogisticRegression(max_iter=25, tol=0.0),
        {"C": loguniform(1e-4, 1e3), "penalty": ["l1", "l2"]},
        n_iter=n_iter, cv=5, random_state=0,
        halving_factor=4, min_resource_frac=0.12,
        return_train_score=False)
    t0 = time.perf_counter()
    rs.fit(X, y)
    dt = time.perf_counter() - t0
    res = rs.cv_results_
    return {"family": "randomized+halving",
            "config": f"{n_iter} candidates, 1Mx256, cv=5, factor 4",
            "wall_s": round(dt, 2),
            "final_rung_candidates": int(np.sum(
                res["resource_iter"] == 25)),
            "best_score": round(rs.best_score_, 4),
            "candidates_per_sec": round(n_iter / dt, 1)}


def bench_sparse():
    """Config 3: mixed-loss grid over a large low-density CSR matrix,
    CSR-resident kernels (K5)."""
    import scipy.sparse as sp

    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import SGDClassifier

    rng = np.random.default_rng(0)
    n, f, k = 10_000_000, 1024, 10          # 1% density, 100M nnz
    indices = rng.integers(0, f, size=(n, k), dtype=np.int32)
    indices.sort(axis=1)
    data = rng.standard_normal(n * k).astype(np.float32)
    indptr = np.arange(0, n * k + 1, k, dtype=np.int32)
    X = sp.csr_matrix((data, indices.ravel(), indptr), shape=(n, f))
    w = rng.standard_normal(f).astype(np.float32)
    y = (np.asarray(X @ w).ravel() > 0).astype(np.int64)

    def one(force_csr):
        import spark_sklearn_amd.ops.sparse_executor as se

        old = se.CSR_MIN_DENSE_BYTES
        se.CSR_MIN_DENSE_BYTES = 0 if force_csr else float("inf")
        try:
            gs = ska.GridSearchCV(
                None, SGDClassifier(max_iter=15, tol=0.0),
                {"alpha": [1e-6, 1e-5, 1e-4, 1e-3],
                 "loss": ["hinge", "log_loss"]}, cv=3,
                return_train_score=False)
            t0 = time.perf_counter()
            gs.fit(X, y)
            return gs, time.perf_counter() - t0
        finally:
            se.CSR_MIN_DENSE_BYTES = old

    gs, dt = one(force_csr=True)            # the K5 CSR-resident kernels
    _, dt_dense = one(force_csr=False)      # dense-ingest comparison point
    return {"family": "sparse-csr", "config": "10Mx1024 CSR d=1% (100M "
            "nnz), mixed hinge/log grid, cv=3, 15 epochs",
            "n_fits": gs.n_fits_, "wall_s": round(dt, 2),
            "dense_ingest_wall_s": round(dt_dense, 2),
            "fits_per_sec": round(gs.n_fits_ / dt, 2),
            "best": gs.best_params_, "best_acc": round(gs.best_score_, 4)}



def bench_rf():
    """The reference's motivating workload shape scaled to 1M rows: a
    RandomForest hyperparameter grid."""
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import RandomForestClassifier

    rng = np.random.default_rng(0)
    n, f = int(os.environ.get("RF_ROWS", "1000000")), 64
    X = rng.standard_normal((n, f), dtype=np.float32)
    y = ((X[:, 0] * X[:, 1] + X[:, 2] + X[:, 3] ** 2 - 1 +
          0.5 * rng.standard_normal(n)) > 0).astype(np.int64)

    grid = {"max_depth": [4, 6], "max_features": [8, 16]}
    gs = ska.GridSearchCV(None,
                          RandomForestClassifier(n_estimators=50,
                                                 random_state=0),
                          grid, cv=3, return_train_score=False)
    t0 = time.perf_counter()
    gs.fit(X, y)
    dt = time.perf_counter() - t0
    return {"family": "rf",
            "config": f"{n}x{f}, 50 trees, depth 4/6 x mf 8/16, cv=3",
            "n_fits": gs.n_fits_, "wall_s": round(dt, 2),
            "fits_per_sec": round(gs.n_fits_ / dt, 3),
            "trees_per_sec": round(gs.n_fits_ * 50 / dt, 1),
            "best": gs.best_params_, "best_acc": round(gs.best_score_, 4)}


def bench_rf_digits():
    """The literal reference README workload: digits + RF grid (~864
    candidates x cv=3 in the original; here the same digits matrix with a
    compact grid plus an 864-candidate variant via RF_DIGITS_FULL=1)."""
    import spark_sklearn_amd as ska
    from sklearn.datasets import load_digits
    from spark_sklearn_amd.models import RandomForestClassifier

    X, y = load_digits(return_X_y=True)
    if os.environ.get("RF_DIGITS_FULL"):
        grid = {"max_depth": [2, 3, 4, 5, 6], "max_features": [2, 4, 6, 8],
                "n_estimators": [20, 40, 80],
                "min_samples_leaf": [1, 2, 4],
                "bootstrap": [True, False]}     # 360 candidates
    else:
        grid = {"max_depth": [4, 6], "max_features": [4, 8],
                "n_estimators": [20, 40]}
    gs = ska.GridSearchCV(None,
                          RandomForestClassifier(random_state=0), grid,
                          cv=3, return_train_score=False)
    t0 = time.perf_counter()
    gs.fit(X, y)
    dt = time.perf_counter() - t0
    return {"family": "rf_digits", "config": "digits 1797x64, cv=3",
            "n_cands": len(gs.cv_results_["params"]),
            "n_fits": gs.n_fits_, "wall_s": round(dt, 2),
            "fits_per_sec": round(gs.n_fits_ / dt, 2),
            "best": gs.best_params_, "best_acc": round(gs.best_score_, 4)}


if __name__ == "__main__":
    which = sys.argv[1:] or ["kmeans", "gbt", "sweep"]
    for name in which:
        fn = {"kmeans": bench_kmeans, "gbt": bench_gbt,
              "sweep": bench_sweep, "sparse": bench_sparse,
              "gbtmulti": bench_gbt_multi, "rf": bench_rf,
              "rfdigits": bench_rf_digits}[name]
        print(json.dumps(fn()), flush=True)
