"""Measure the CSR-vs-dense crossover density (VERDICT round-1 item 6).

Runs the same mixed linear grid through the CSR-resident executor and the
dense-ingest executor at several densities; prints a JSON line per cell.
GPU box:  python benchmarks/diag/density_crossover.py
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import numpy as np
import scipy.sparse as sp

import spark_sklearn_amd as ska
from spark_sklearn_amd.models import LogisticRegression


def run(X, y, label):
    # 64 candidates: amortizes one-time prep (CSR transpose argsort /
    # dense ingest) so the cell measures steady-state per-epoch cost
    gs = ska.GridSearchCV(None, LogisticRegression(max_iter=15, tol=0.0),
                          {"C": list(np.logspace(-3, 3, 64))}, cv=3)
    t0 = time.perf_counter()
    gs.fit(X, y)
    dt = time.perf_counter() - t0
    return {"path": label, "wall_s": round(dt, 3),
            "best": round(gs.best_score_, 4)}


def main():
    rng = np.random.default_rng(0)
    n, f = 2_000_000, 512
    for dens in (0.005, 0.01, 0.02, 0.05, 0.10):
        Xs = sp.random(n, f, density=dens, format="csr", random_state=1,
                       dtype=np.float32)
        w = rng.standard_normal(f).astype(np.float32)
        y = (np.asarray(Xs @ w).ravel() +
             0.2 * rng.standard_normal(n) > 0).astype(np.int64)
        import spark_sklearn_amd.ops.sparse_executor as se
        from spark_sklearn_amd.ops import sparse_executor

        # force each path regardless of the 5% auto-gate
        old_limit = se.CSR_DENSITY_LIMIT
        import spark_sklearn_amd.engine as eng
        se.CSR_DENSITY_LIMIT = 1.0
        eng.CSR_DENSITY_LIMIT = 1.0
        r_csr = run(Xs, y, "csr")
        se.CSR_DENSITY_LIMIT = 0.0
        eng.CSR_DENSITY_LIMIT = 0.0
        r_dense = run(Xs, y, "dense-ingest")
        se.CSR_DENSITY_LIMIT = old_limit
        eng.CSR_DENSITY_LIMIT = old_limit
        print(json.dumps({"density": dens, "csr_s": r_csr["wall_s"],
                          "dense_s": r_dense["wall_s"],
                          "csr_faster": r_csr["wall_s"] < r_dense["wall_s"]}),
              flush=True)


if __name__ == "__main__":
    main()
